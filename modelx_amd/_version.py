"""Version stamp (reference: pkg/version/version.go:9-38).

The Makefile stamps ``GIT_VERSION``/``GIT_COMMIT`` at build time by writing
``modelx_amd/_build_stamp.py``; absent that, values fall back to dev defaults.
"""
from __future__ import annotations

import platform
from dataclasses import asdict, dataclass

GIT_VERSION = "v0.1.0-dev"
GIT_COMMIT = "unknown"
BUILD_DATE = "unknown"

try:  # written by `make stamp` / setup build
    from ._build_stamp import BUILD_DATE, GIT_COMMIT, GIT_VERSION  # type: ignore # noqa: F811,F401
except ImportError:
    pass

__version__ = GIT_VERSION


@dataclass
class Version:
    gitVersion: str = GIT_VERSION
    gitCommit: str = GIT_COMMIT
    buildDate: str = BUILD_DATE
    goVersion: str = ""  # kept for wire parity with the reference's Get()
    compiler: str = "hipcc/amdclang++"
    platform: str = f"{platform.system().lower()}/{platform.machine()}"

    def to_dict(self):
        return asdict(self)


def get() -> Version:
    return Version()
