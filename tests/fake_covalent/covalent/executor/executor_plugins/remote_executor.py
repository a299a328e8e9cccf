class RemoteExecutor:
    """Shape of covalent's RemoteExecutor template (ctor stores
    poll_freq/remote_cache; dispatcher awaits run())."""

    def __init__(self, poll_freq=15, remote_cache="", *args, **kwargs):
        self.poll_freq = poll_freq
        self.remote_cache = remote_cache
