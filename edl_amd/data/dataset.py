"""File splitting (parity: reference collective/dataset.py:16-44
FileSplitter/TxtFileSplitter — the user extension point that turns files
into (record_idx, data) streams)."""


class FileSplitter:
    """Subclass and implement split(path) -> iterable of records."""

    def split(self, path):
        raise NotImplementedError

    def __call__(self, path):
        for i, rec in enumerate(self.split(path)):
            yield i, rec


class TxtFileSplitter(FileSplitter):
    """One record per line (stripped)."""

    def split(self, path):
        with open(path, "r") as f:
            for line in f:
                line = line.rstrip("\n")
                if line:
                    yield line
