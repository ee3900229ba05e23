"""Connectors for local text files.

Parity target: ``bytewax.connectors.files`` (reference
connectors/files.py:37-463): per-file partitions keyed
``fs_id::path``, byte-offset snapshots (`tell`/`seek`) for
exactly-once resume, sinks that `seek`+`truncate` back to the
snapshot offset on resume and fsync per batch.
"""

import os
from csv import DictReader
from pathlib import Path
from typing import Callable, Dict, Iterator, List, Optional, Union

from ..inputs import FixedPartitionedSource, StatefulSourcePartition, batch
from ..outputs import FixedPartitionedSink, StatefulSinkPartition

__all__ = [
    "CSVSource",
    "DirSink",
    "DirSource",
    "FileSink",
    "FileSource",
]


def _get_path_dev(path: Path) -> str:
    return hex(path.stat().st_dev)


def _readlines(f) -> Iterator[str]:
    # `next(file)` disables `tell`; re-create the iterator with
    # `readline` so offsets stay queryable.
    while True:
        line = f.readline()
        if len(line) <= 0:
            break
        yield line


def _strip_n(s: str) -> str:
    return s.rstrip("\n")


class _FileSourcePartition(StatefulSourcePartition[str, int]):
    def __init__(self, path: Path, batch_size: int, resume_state: Optional[int]):
        self._f = open(path, "rt")
        if resume_state is not None:
            self._f.seek(resume_state)
        self._batcher = batch(map(_strip_n, _readlines(self._f)), batch_size)

    def next_batch(self) -> List[str]:
        return next(self._batcher)

    def snapshot(self) -> int:
        return self._f.tell()

    def close(self) -> None:
        self._f.close()


class DirSource(FixedPartitionedSource[str, int]):
    """Read all files in a filesystem directory line-by-line.

    Unique files are the unit of parallelism; only one worker reads
    each unique file, so lines from different files are interleaved.
    Supports exactly-once processing.

    :arg dir_path: Path to directory.
    :arg glob_pat: Pattern of files to read; defaults to `"*"`.
    :arg batch_size: Lines per batch; defaults to 1000.
    :arg get_fs_id: Returns a consistent unique ID for the filesystem
        of the directory (defaults to `st_dev`); return a constant if
        all workers see identical files.
    """

    def __init__(
        self,
        dir_path: Path,
        glob_pat: str = "*",
        batch_size: int = 1000,
        get_fs_id: Callable[[Path], str] = _get_path_dev,
    ):
        dir_path = Path(dir_path)
        if not dir_path.exists():
            msg = f"input directory `{dir_path}` does not exist"
            raise ValueError(msg)
        if not dir_path.is_dir():
            msg = f"input directory `{dir_path}` is not a directory"
            raise ValueError(msg)
        self._dir_path = dir_path
        self._glob_pat = glob_pat
        self._batch_size = batch_size
        self._fs_id = get_fs_id(dir_path)
        if "::" in self._fs_id:
            msg = (
                "result of `get_fs_id` must not contain `::`; "
                f"got {self._fs_id!r}"
            )
            raise ValueError(msg)

    def list_parts(self) -> List[str]:
        if not self._dir_path.exists():
            return []
        return [
            f"{self._fs_id}::{path}"
            for path in sorted(self._dir_path.glob(self._glob_pat))
            if path.is_file()
        ]

    def build_part(
        self, step_id: str, for_part: str, resume_state: Optional[int]
    ) -> _FileSourcePartition:
        _fs_id, path = for_part.split("::", 1)
        return _FileSourcePartition(Path(path), self._batch_size, resume_state)


class FileSource(FixedPartitionedSource[str, int]):
    """Read a single file line-by-line from the filesystem.

    This file must exist and be identical on all workers.  There is no
    parallelism; only one worker actually reads the file.
    """

    def __init__(
        self,
        path: Union[Path, str],
        batch_size: int = 1000,
        get_fs_id: Callable[[Path], str] = _get_path_dev,
    ):
        self._path = Path(path)
        self._batch_size = batch_size
        self._get_fs_id = get_fs_id

    def list_parts(self) -> List[str]:
        return [f"{self._get_fs_id(self._path.parent)}::{self._path}"]

    def build_part(
        self, step_id: str, for_part: str, resume_state: Optional[int]
    ) -> _FileSourcePartition:
        _fs_id, path = for_part.split("::", 1)
        return _FileSourcePartition(Path(path), self._batch_size, resume_state)


class _CSVPartition(StatefulSourcePartition[Dict[str, str], int]):
    def __init__(
        self, path: Path, batch_size: int, resume_state: Optional[int], fmtparams
    ):
        self._f = open(path, "rt", newline="")
        # The header must be read from the start to know field names.
        header = self._f.readline().rstrip("\r\n")
        import csv as _csv

        dialect_kwargs = dict(fmtparams)
        delimiter = dialect_kwargs.get("delimiter", ",")
        self._fields = next(_csv.reader([header], **fmtparams), [])
        if resume_state is not None:
            self._f.seek(resume_state)
        self._reader = DictReader(
            _readlines(self._f), fieldnames=self._fields, **fmtparams
        )
        self._batcher = batch(self._reader, batch_size)

    def next_batch(self) -> List[Dict[str, str]]:
        return next(self._batcher)

    def snapshot(self) -> int:
        return self._f.tell()

    def close(self) -> None:
        self._f.close()


class CSVSource(FixedPartitionedSource[Dict[str, str], int]):
    """Read a single CSV file row-by-row as keyed dictionaries.

    The file must exist and be identical on all workers.  The first
    row is the header.
    """

    def __init__(
        self,
        path: Path,
        batch_size: int = 1000,
        get_fs_id: Callable[[Path], str] = _get_path_dev,
        **fmtparams,
    ):
        self._path = Path(path)
        self._batch_size = batch_size
        self._get_fs_id = get_fs_id
        self._fmtparams = fmtparams

    def list_parts(self) -> List[str]:
        return [f"{self._get_fs_id(self._path.parent)}::{self._path}"]

    def build_part(
        self, step_id: str, for_part: str, resume_state: Optional[int]
    ) -> _CSVPartition:
        _fs_id, path = for_part.split("::", 1)
        return _CSVPartition(
            Path(path), self._batch_size, resume_state, self._fmtparams
        )


class _FileSinkPartition(StatefulSinkPartition[str, int]):
    def __init__(self, path: Path, resume_state: Optional[int], end: str):
        resume_offset = 0 if resume_state is None else resume_state
        path.parent.mkdir(parents=True, exist_ok=True)
        self._f = open(path, "at")
        # Truncate back to the last snapshot offset: output written
        # after the last completed epoch is rolled back, giving
        # exactly-once output in a batch context.
        self._f.seek(resume_offset)
        self._f.truncate()
        self._end = end

    def write_batch(self, values: List[str]) -> None:
        for v in values:
            self._f.write(v)
            self._f.write(self._end)
        self._f.flush()
        os.fsync(self._f.fileno())

    def snapshot(self) -> int:
        return self._f.tell()

    def close(self) -> None:
        self._f.close()


class DirSink(FixedPartitionedSink[str, int]):
    """Write to a set of files in a filesystem directory line-by-line.

    Items consumed from the dataflow must be `(key, value)` 2-tuples;
    the value is written, the key routes to one of `file_count` files
    named by `file_namer`.  Supports exactly-once processing in a
    batch context via offset truncation on resume.
    """

    def __init__(
        self,
        dir_path: Path,
        file_count: int,
        file_namer: Callable[[int, int], str] = None,
        assign_file: Callable[[str], int] = None,
        end: str = "\n",
    ):
        dir_path = Path(dir_path)
        if not dir_path.exists():
            msg = f"output directory `{dir_path}` does not exist"
            raise ValueError(msg)
        self._dir_path = dir_path
        self._file_count = file_count
        # Default name matches the reference ("part_{i}", no
        # extension; reference files.py:372).
        self._file_namer = file_namer or (lambda i, n: f"part_{i}")
        self._assign_file = assign_file
        self._end = end

    def list_parts(self) -> List[str]:
        return [
            self._file_namer(i, self._file_count)
            for i in range(self._file_count)
        ]

    def part_fn(self, item_key: str) -> int:
        if self._assign_file is not None:
            return self._assign_file(item_key)
        return super().part_fn(item_key)

    def build_part(
        self, step_id: str, for_part: str, resume_state: Optional[int]
    ) -> _FileSinkPartition:
        return _FileSinkPartition(
            self._dir_path / for_part, resume_state, self._end
        )


class FileSink(FixedPartitionedSink[str, int]):
    """Write to a single file line-by-line on the filesystem.

    Items consumed from the dataflow must be `(key, value)` 2-tuples.
    The file will be created on one worker; supports exactly-once
    processing in a batch context.
    """

    def __init__(self, path: Path, end: str = "\n"):
        self._path = Path(path)
        self._end = end

    def list_parts(self) -> List[str]:
        return [str(self._path)]

    def build_part(
        self, step_id: str, for_part: str, resume_state: Optional[int]
    ) -> _FileSinkPartition:
        return _FileSinkPartition(Path(for_part), resume_state, self._end)
