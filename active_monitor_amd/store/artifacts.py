"""Artifact readers (reference: internal/store/{store,inline,url}.go).

Error-message compatibility notes (pinned by the reference's tests,
store_test.go:59-66):

- unknown source      → ``unknown artifact location: <loc>``
- empty inline        → ``InlineArtifact does not exist``
- missing URLArtifact → ``URLArtifact cannot be empty``
- non-200 HTTP status → ``status code <n>``
"""
from __future__ import annotations

import logging
import ssl
import urllib.error
import urllib.request
from pathlib import Path
from typing import Optional

from ..api.types import ArtifactLocation, URLArtifact

log = logging.getLogger("active_monitor_amd.store")


class ArtifactReadError(Exception):
    """Raised when an artifact source cannot be constructed or read."""


class ArtifactReader:
    """Reads a workflow definition from an external store
    (interface parity: store.go:10-12)."""

    def read(self) -> bytes:  # pragma: no cover - interface
        raise NotImplementedError


class InlineReader(ArtifactReader):
    """Returns the inline YAML string (inline.go:15-26)."""

    def __init__(self, inline: Optional[str]):
        if inline is None or inline == "":
            raise ArtifactReadError("InlineArtifact does not exist")
        self._inline = inline

    def read(self) -> bytes:
        log.debug("reading fileArtifact from inline")
        return self._inline.encode("utf-8")


#: hard ceiling on one artifact fetch; the reference's Go http.Client has no
#: timeout but blocks only its own goroutine — here a bounded read keeps a
#: hung source from pinning a worker thread forever
DEFAULT_READ_TIMEOUT = 30.0


class URLReader(ArtifactReader):
    """HTTP GET of the workflow YAML; TLS verification on unless
    ``verifyCert: false`` (url.go:20-57, secure by default)."""

    def __init__(self, url: Optional[URLArtifact], timeout: float = DEFAULT_READ_TIMEOUT):
        if url is None:
            raise ArtifactReadError("URLArtifact cannot be empty")
        self._url = url
        self._timeout = timeout

    def read(self) -> bytes:
        log.debug("reading urlArtifact from %s", self._url.path)
        ctx = None
        if not self._url.should_verify:
            log.warning("TLS certificate verification is disabled for %s", self._url.path)
            ctx = ssl.create_default_context()
            ctx.check_hostname = False
            ctx.verify_mode = ssl.CERT_NONE
        try:
            with urllib.request.urlopen(self._url.path, context=ctx, timeout=self._timeout) as resp:
                status = getattr(resp, "status", 200)
                if status != 200:
                    raise ArtifactReadError(f"status code {status}")
                return resp.read()
        except urllib.error.HTTPError as e:
            log.warning("failed to read %s. status code: %d", self._url.path, e.code)
            raise ArtifactReadError(f"status code {e.code}") from e
        except urllib.error.URLError as e:
            log.warning("failed to read url %s: %s", self._url.path, e)
            raise ArtifactReadError(str(e)) from e
        except OSError as e:  # read-side socket timeout / reset
            log.warning("failed to read url %s: %s", self._url.path, e)
            raise ArtifactReadError(f"failed to read {self._url.path}: {e}") from e


class FileReader(ArtifactReader):
    """Reads the workflow YAML from a local file path.

    The reference API declares this source but its store never implements it
    (SURVEY.md §2.3.8); implemented here to make the documented API real."""

    def __init__(self, path: Optional[str]):
        if not path:
            raise ArtifactReadError("FileArtifact does not exist")
        self._path = Path(path)

    def read(self) -> bytes:
        log.debug("reading fileArtifact from %s", self._path)
        try:
            return self._path.read_bytes()
        except OSError as e:
            raise ArtifactReadError(f"failed to read file {self._path}: {e}") from e


def get_artifact_reader(loc: Optional[ArtifactLocation], allow_file: bool = True) -> ArtifactReader:
    """Factory dispatch (store.go:15-22). Order matches the reference:
    inline first, then URL; with ``allow_file=False`` a ``file`` source falls
    through to the reference's ``unknown artifact location`` error."""
    if loc is not None:
        if loc.inline is not None:
            return InlineReader(loc.inline)
        if loc.url is not None:
            return URLReader(loc.url)
        if allow_file and loc.file is not None:
            return FileReader(loc.file.path)
    shown = loc.to_dict() if loc is not None else None
    raise ArtifactReadError(f"unknown artifact location: {shown}")
