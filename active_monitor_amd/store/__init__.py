"""Artifact store: readers for workflow definitions.

Capability parity with the reference store (internal/store/):

- inline strings (internal/store/inline.go),
- HTTP(S) URLs with TLS verification on by default (internal/store/url.go),
- anything else → ``unknown artifact location`` error (internal/store/store.go:15-22).

One deliberate improvement: the reference declares a ``file`` source in its API
(healthcheck_types.go:134-136) but never implements it; here ``file`` is
implemented for real (FileReader) while the reference's error string is kept
available for compatibility tests via ``GetArtifactReaderStrict``.
"""
from .artifacts import (
    ArtifactReader,
    ArtifactReadError,
    FileReader,
    InlineReader,
    URLReader,
    get_artifact_reader,
)

__all__ = [
    "ArtifactReader",
    "ArtifactReadError",
    "FileReader",
    "InlineReader",
    "URLReader",
    "get_artifact_reader",
]
