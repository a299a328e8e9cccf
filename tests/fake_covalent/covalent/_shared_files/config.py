_config = {
    "dispatcher.cache_dir": "/tmp/fake-covalent-cache",
    "executors.ssh.username": "cfg-user",
    "executors.ssh.hostname": "cfg-host",
    "executors.ssh.python_path": "python3",
}


def get_config(key):
    return _config[key]  # KeyError on unknown keys, like the real one


def set_config(key, value=None):
    if isinstance(key, dict):
        _config.update(key)
    else:
        _config[key] = value
