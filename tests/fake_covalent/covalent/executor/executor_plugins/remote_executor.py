"""Transcription of covalent's RemoteExecutor public contract.

Reconstructed (no covalent wheel exists offline) from three corroborating
sources:

* the reference plugin's observable usage: ``super().__init__(
  poll_freq=poll_freq, remote_cache=remote_cache)`` at reference
  ssh.py:98, and its implementation of exactly the seven template
  methods at reference ssh.py:317, 337, 363, 388, 408, 434, 460;
* reference CHANGELOG.md:110-118 ("Using `RemoteExecutor` now instead of
  `BaseAsyncExecutor` ... Implementation of abstract functions added to
  adhere to the `RemoteExecutor`'s template");
* covalent 0.2xx's documented executor API: RemoteExecutor(poll_freq,
  remote_cache, credentials_file) extending AsyncBaseExecutor(log_stdout,
  log_stderr, cache_dir, time_limit, retries), whose dispatcher awaits
  ``run(function, args, kwargs, task_metadata)``.

tests/test_covalent_integration.py pins SSHExecutor against THIS
signature set, so any divergence between our compat shim and the
documented contract fails loudly.
"""


class AsyncBaseExecutor:
    """Base async executor (covalent/executor/base.py shape)."""

    def __init__(
        self,
        log_stdout: str = "",
        log_stderr: str = "",
        cache_dir: str = "",
        time_limit: int = -1,
        retries: int = 0,
        *args,
        **kwargs,
    ):
        self.log_stdout = log_stdout
        self.log_stderr = log_stderr
        self.cache_dir = cache_dir
        self.time_limit = time_limit
        self.retries = retries

    async def run(self, function, args, kwargs, task_metadata):
        raise NotImplementedError


class RemoteExecutor(AsyncBaseExecutor):
    """Template for executors that run tasks on remote machines."""

    def __init__(
        self,
        poll_freq: int = 15,
        remote_cache: str = "",
        credentials_file: str = "",
        *args,
        **kwargs,
    ):
        super().__init__(*args, **kwargs)
        self.poll_freq = poll_freq
        self.remote_cache = remote_cache
        self.credentials_file = credentials_file

    # --- the seven abstract template methods the plugin must implement
    async def _validate_credentials(self, *args, **kwargs):
        raise NotImplementedError

    async def _upload_task(self, *args, **kwargs):
        raise NotImplementedError

    async def submit_task(self, *args, **kwargs):
        raise NotImplementedError

    async def get_status(self, *args, **kwargs):
        raise NotImplementedError

    async def _poll_task(self, *args, **kwargs):
        raise NotImplementedError

    async def query_result(self, *args, **kwargs):
        raise NotImplementedError

    async def cancel(self, *args, **kwargs):
        raise NotImplementedError
