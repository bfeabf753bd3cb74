"""Input taps: walk files/dirs/globs and chunk them into byte-range datasets.

Role parity with the reference's ``dampr/inputs.py`` (reference:
inputs.py:14-98).
"""
import glob
import os

from .dataset import (Chunker, Dataset, GzipLineDataset, MemoryDataset,
                      TextLineDataset)


def read_paths(paths, follow_links=False):
    """Expand globs/dirs into file paths, skipping dotfiles
    (reference: inputs.py:14-30)."""
    if not isinstance(paths, list):
        paths = [paths]

    def walk():
        for pattern in paths:
            for path in glob.glob(pattern):
                if os.path.isfile(path):
                    yield path
                else:
                    for root, _dirs, files in os.walk(
                            path, followlinks=follow_links):
                        for fname in files:
                            yield os.path.join(root, fname)

    return (p for p in walk()
            if not os.path.basename(p).startswith("."))


class TextInput(Chunker):
    """One file chunked into byte ranges (gzip: whole file)."""

    def __init__(self, path, chunk_size=64 * 1024 ** 2):
        self.path = path
        self.chunk_size = chunk_size

    def chunks(self):
        if self.path.endswith(".gz"):
            yield GzipLineDataset(self.path)
            return
        size = os.stat(self.path).st_size
        offset = 0
        while offset < size or offset == 0 == size:
            yield TextLineDataset(self.path, offset,
                                  offset + self.chunk_size)
            offset += self.chunk_size
            if size == 0:
                break


class PathInput(Chunker):
    """Files, directories and globs."""

    def __init__(self, path, chunk_size=64 * 1024 ** 2, follow_links=True):
        self.path = path
        self.chunk_size = chunk_size
        self.follow_links = follow_links

    def chunks(self):
        for path in read_paths(self.path, self.follow_links):
            for chunk in TextInput(path, self.chunk_size).chunks():
                yield chunk


class MemoryInput(Chunker):
    """An in-memory list of (k, v), split into ``partitions`` chunks."""

    def __init__(self, items, partitions=50):
        self.items = items
        self.partitions = min(len(items), partitions)

    def chunks(self):
        if self.partitions == 0:
            yield MemoryDataset(self.items)
            return
        chunk_size = max(1, len(self.items) // self.partitions)
        for start in range(0, len(self.items), chunk_size):
            yield MemoryDataset(self.items[start:start + chunk_size])


class UrlDataset(Dataset):
    """Streams lines from a URL (kept for API parity; this build
    environment has no egress)."""

    def __init__(self, path, skip_on_error=True):
        self.path = path
        self.soe = skip_on_error

    def read(self):
        from urllib.request import urlopen
        from urllib.error import URLError
        try:
            with urlopen(self.path) as h:
                for i, line in enumerate(h):
                    yield i, line.decode("utf-8")
        except URLError:
            if not self.soe:
                raise


class UrlsInput(Chunker):
    def __init__(self, urls, skip_on_error=True):
        self.urls = urls
        self.soe = skip_on_error

    def chunks(self):
        for url in self.urls:
            yield UrlDataset(url, self.soe)
