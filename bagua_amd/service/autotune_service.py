"""Autotune service (placeholder — HTTP service lands with the autotune commit)."""


def start_autotune_server(port, world_size):
    raise NotImplementedError


class AutotuneClient:
    def __init__(self, host, port):
        raise NotImplementedError
