"""IO subsystem: URI-schemed byte streams + buffered text reading.

Capability parity with the reference io layer (include/multiverso/io/io.h:
24-117, src/io/local_stream.cpp, src/io/io.cpp:25-58): ``URI`` scheme
parsing, an abstract ``Stream`` (Write/Read/Good), a per-scheme
``StreamFactory``, a local-file implementation, and ``TextReader``'s
buffered get_line. HDFS (the reference's optional libhdfs backend) is
registered as an explicit unavailable scheme — constructing it raises with
the same capability boundary the reference has when built without
MULTIVERSO_USE_HDFS."""

from __future__ import annotations

from typing import Callable, Dict, Optional


class URI:
    def __init__(self, uri: str) -> None:
        if "://" in uri:
            self.scheme, self.path = uri.split("://", 1)
        else:
            self.scheme, self.path = "file", uri
        self.raw = uri


class Stream:
    def write(self, data: bytes) -> int:
        raise NotImplementedError

    def read(self, size: int = -1) -> bytes:
        raise NotImplementedError

    def good(self) -> bool:
        raise NotImplementedError

    def close(self) -> None:
        pass

    def __enter__(self) -> "Stream":
        return self

    def __exit__(self, *exc) -> None:
        self.close()


class LocalStream(Stream):
    def __init__(self, uri: URI, mode: str) -> None:
        if "b" not in mode:
            mode += "b"
        self._f = open(uri.path, mode)
        self._good = True

    def write(self, data: bytes) -> int:
        self._f.write(data)
        return len(data)

    def read(self, size: int = -1) -> bytes:
        return self._f.read(size)

    def good(self) -> bool:
        return self._good and not self._f.closed

    def close(self) -> None:
        self._f.close()


def _hdfs_unavailable(uri: URI, mode: str) -> Stream:
    raise NotImplementedError(
        "hdfs:// streams require libhdfs (reference MULTIVERSO_USE_HDFS "
        "build option); not available in this environment")


class StreamFactory:
    _registry: Dict[str, Callable[[URI, str], Stream]] = {
        "file": LocalStream,
        "hdfs": _hdfs_unavailable,
    }

    @classmethod
    def register(cls, scheme: str,
                 ctor: Callable[[URI, str], Stream]) -> None:
        cls._registry[scheme] = ctor

    @classmethod
    def get_stream(cls, uri, mode: str = "r") -> Stream:
        u = uri if isinstance(uri, URI) else URI(uri)
        try:
            ctor = cls._registry[u.scheme]
        except KeyError:
            raise ValueError(f"no stream backend for scheme '{u.scheme}'")
        return ctor(u, mode)


class TextReader:
    """Buffered line reader over a Stream (reference io.cpp:25-58)."""

    def __init__(self, uri, buf_size: int = 1 << 16) -> None:
        self._stream = StreamFactory.get_stream(uri, "r")
        self._buf = b""
        self._buf_size = buf_size
        self._eof = False

    def get_line(self) -> Optional[str]:
        while b"\n" not in self._buf and not self._eof:
            chunk = self._stream.read(self._buf_size)
            if not chunk:
                self._eof = True
                break
            self._buf += chunk
        if b"\n" in self._buf:
            line, self._buf = self._buf.split(b"\n", 1)
            return line.decode()
        if self._buf:
            line, self._buf = self._buf, b""
            return line.decode()
        return None

    def close(self) -> None:
        self._stream.close()
