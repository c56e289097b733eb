"""Storage abstraction: local disk or HDFS (via the ``hdfs dfs`` CLI).

Mirrors the reference ``persia-storage`` crate (persia-storage/src/lib.rs:13-391):
``PersiaPath`` dispatches on an ``hdfs://`` prefix; HDFS operations shell out
to the hadoop CLI exactly like the reference does (``-put``, ``-text``,
``-ls``, ``-rm``, ``-appendToFile``, ``-touchz``)."""
import os
import subprocess
import tempfile
from typing import List


def is_hdfs(path: str) -> bool:
    return path.startswith("hdfs://")


class PersiaPath:
    def __init__(self, path: str):
        self.path = path
        self.hdfs = is_hdfs(path)

    def _hdfs(self, *args: str, input_bytes=None) -> bytes:
        cmd = ["hdfs", "dfs", *args]
        out = subprocess.run(cmd, input=input_bytes, capture_output=True, check=True)
        return out.stdout

    def create(self) -> None:
        if self.hdfs:
            self._hdfs("-mkdir", "-p", os.path.dirname(self.path))
            self._hdfs("-touchz", self.path)
        else:
            os.makedirs(os.path.dirname(os.path.abspath(self.path)), exist_ok=True)
            open(self.path, "ab").close()

    def is_file(self) -> bool:
        if self.hdfs:
            try:
                self._hdfs("-test", "-e", self.path)
                return True
            except subprocess.CalledProcessError:
                return False
        return os.path.isfile(self.path)

    def read_to_end(self) -> bytes:
        if self.hdfs:
            return self._hdfs("-text", self.path)
        with open(self.path, "rb") as f:
            return f.read()

    def write_all(self, data: bytes) -> None:
        if self.hdfs:
            with tempfile.NamedTemporaryFile() as tmp:
                tmp.write(data)
                tmp.flush()
                self._hdfs("-put", "-f", tmp.name, self.path)
        else:
            os.makedirs(os.path.dirname(os.path.abspath(self.path)), exist_ok=True)
            with open(self.path, "wb") as f:
                f.write(data)

    def append(self, data: bytes) -> None:
        if self.hdfs:
            self._hdfs("-appendToFile", "-", self.path, input_bytes=data)
        else:
            with open(self.path, "ab") as f:
                f.write(data)

    def list(self) -> List[str]:
        if self.hdfs:
            out = self._hdfs("-ls", self.path).decode()
            return [line.split()[-1] for line in out.splitlines() if line.startswith("-") or line.startswith("d")]
        if not os.path.isdir(self.path):
            return []
        return [os.path.join(self.path, p) for p in sorted(os.listdir(self.path))]

    def remove(self) -> None:
        if self.hdfs:
            self._hdfs("-rm", "-r", "-f", self.path)
        elif os.path.isfile(self.path):
            os.remove(self.path)
