"""Storage abstraction (Java API parity: fileio/RapidsFileIO.java:28-60 —
input/output file + seekable stream interfaces so the execution layer can
route reads to HDFS/S3/local without touching the scan code)."""
import io
import os
from abc import ABC, abstractmethod
from typing import BinaryIO


class SeekableInputStream(ABC):
    @abstractmethod
    def read(self, n: int = -1) -> bytes: ...

    @abstractmethod
    def seek(self, pos: int) -> None: ...

    @abstractmethod
    def tell(self) -> int: ...

    def read_fully(self, pos: int, n: int) -> bytes:
        """Exactly n bytes at pos: EOFError on a short read, ValueError on
        negative arguments (reference RapidsInputFileTest semantics)."""
        if pos < 0 or n < 0:
            raise ValueError(f"negative read: pos={pos} n={n}")
        self.seek(pos)
        out = self.read(n)
        if len(out) != n:
            raise EOFError(f"short read at {pos}: {len(out)} != {n}")
        return out


class InputFile(ABC):
    @abstractmethod
    def length(self) -> int: ...

    @abstractmethod
    def open(self) -> SeekableInputStream: ...

    def read_tail(self, n: int) -> bytes:
        """Last n bytes of the file (reference RapidsInputFile.readTail):
        EOFError when n exceeds the file length — footer readers rely on
        this to fail cleanly on truncated files."""
        ln = self.length()
        if n > ln:
            raise EOFError(f"readTail({n}) > file length {ln}")
        return self.open().read_fully(ln - n, n)


class OutputFile(ABC):
    @abstractmethod
    def create(self, overwrite: bool = False) -> BinaryIO: ...


class RapidsFileIO(ABC):
    @abstractmethod
    def new_input_file(self, path: str) -> InputFile: ...

    @abstractmethod
    def new_output_file(self, path: str) -> OutputFile: ...


class _LocalStream(SeekableInputStream):
    def __init__(self, f):
        self._f = f

    def read(self, n=-1):
        return self._f.read(n)

    def seek(self, pos):
        self._f.seek(pos)

    def tell(self):
        return self._f.tell()


class LocalInputFile(InputFile):
    def __init__(self, path: str):
        self.path = path

    def length(self) -> int:
        return os.path.getsize(self.path)

    def open(self) -> SeekableInputStream:
        return _LocalStream(open(self.path, "rb"))


class LocalOutputFile(OutputFile):
    def __init__(self, path: str):
        self.path = path

    def create(self, overwrite=False) -> BinaryIO:
        if not overwrite and os.path.exists(self.path):
            raise FileExistsError(self.path)
        return open(self.path, "wb")


class LocalFileIO(RapidsFileIO):
    def new_input_file(self, path: str) -> InputFile:
        return LocalInputFile(path)

    def new_output_file(self, path: str) -> OutputFile:
        return LocalOutputFile(path)
