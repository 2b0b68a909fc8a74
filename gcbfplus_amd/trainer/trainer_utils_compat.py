"""tqdm shim: use the real tqdm if present, else a minimal stand-in."""
try:
    from tqdm import tqdm  # noqa: F401
except ImportError:  # pragma: no cover

    class tqdm:  # type: ignore
        def __init__(self, total=None, ncols=None, **kw):
            self.total = total
            self.n = 0

        def update(self, k=1):
            self.n += k

        def close(self):
            pass

        @staticmethod
        def write(msg):
            print(msg)
