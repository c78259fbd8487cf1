"""Multi-progress-bar transfer scheduler
(reference: pkg/client/progress/{mbar,bar,bar-io}.go).

Keeps the reference's model: a MultiBar owns a worker pool with a concurrency
limit (the transfer scheduler, mbar.go:95-120), each transfer owns a Bar, and
a Bar is made of *fragments* (per-part progress, bar.go:15-35) — fragments map
1:1 onto the GPU chunk-pipeline stages. A failure cancels the siblings
(mbar.go:110-114). Rendering is a 100 ms ticker on a TTY, silent otherwise.
"""
from __future__ import annotations

import sys
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from typing import Callable, List, Optional

from .units import human_size


class Fragment:
    __slots__ = ("offset", "total", "done")

    def __init__(self, offset: int = 0, total: int = 0):
        self.offset = offset
        self.total = total
        self.done = 0


class Bar:
    def __init__(self, name: str, total: int = 0, status: str = ""):
        self.name = name
        self.total = total
        self.status = status
        self.fragments: List[Fragment] = []
        self.failed = False
        self.complete = False
        self._lock = threading.Lock()
        self._start = time.monotonic()

    def add_fragment(self, offset: int, total: int) -> Fragment:
        f = Fragment(offset, total)
        with self._lock:
            self.fragments.append(f)
        return f

    def advance(self, n: int, fragment: Optional[Fragment] = None) -> None:
        with self._lock:
            if fragment is None:
                if not self.fragments:
                    self.fragments.append(Fragment(0, self.total))
                fragment = self.fragments[0]
            fragment.done += n

    @property
    def done_bytes(self) -> int:
        return sum(f.done for f in self.fragments)

    def set_status(self, status: str, complete: bool = False, failed: bool = False) -> None:
        self.status = status
        self.complete = complete
        self.failed = failed

    def rate(self) -> float:
        dt = time.monotonic() - self._start
        return self.done_bytes / dt if dt > 0 else 0.0

    def render(self, width: int = 60) -> str:
        done = self.done_bytes
        pct = min(1.0, done / self.total) if self.total else (1.0 if self.complete else 0.0)
        nfill = int(pct * 20)
        bar = "[" + "#" * nfill + "-" * (20 - nfill) + "]"
        name = self.name if len(self.name) <= 24 else self.name[:21] + "..."
        tail = self.status or f"{human_size(done)}/{human_size(self.total)} {human_size(self.rate())}/s"
        return f"{name:<24} {bar} {tail}"


class MultiBar:
    """Worker pool + renderer. ``concurrency`` mirrors PullPushConcurrency=3
    (push.go:27) as the default; the GPU engine raises it."""

    def __init__(self, description: str = "", concurrency: int = 3, quiet: Optional[bool] = None):
        self.description = description
        self.bars: List[Bar] = []
        self._lock = threading.Lock()
        self._pool = ThreadPoolExecutor(max_workers=max(1, concurrency))
        self._futures = []
        self._cancelled = threading.Event()
        self._quiet = (not sys.stderr.isatty()) if quiet is None else quiet
        self._render_thread: Optional[threading.Thread] = None
        self._stop_render = threading.Event()
        self._lines_drawn = 0

    def new_bar(self, name: str, total: int = 0) -> Bar:
        bar = Bar(name, total)
        with self._lock:
            self.bars.append(bar)
        return bar

    def cancelled(self) -> bool:
        return self._cancelled.is_set()

    def go(self, name: str, total: int, fn: Callable[[Bar], None]) -> None:
        bar = self.new_bar(name, total)

        def run():
            if self._cancelled.is_set():
                bar.set_status("cancelled", failed=True)
                return
            try:
                fn(bar)
                if not bar.failed and not bar.complete:
                    bar.set_status("done", complete=True)
            except BaseException as e:
                bar.set_status(f"failed: {e}", failed=True)
                self._cancelled.set()  # cancel siblings (mbar.go:110-114)
                raise

        self._futures.append(self._pool.submit(run))

    def _render_loop(self):
        while not self._stop_render.wait(0.1):
            self._render_once()

    def _render_once(self):
        with self._lock:
            lines = [b.render() for b in self.bars]
        out = ""
        if self._lines_drawn:
            out += f"\x1b[{self._lines_drawn}A"  # cursor up
        for ln in lines:
            out += "\x1b[2K" + ln + "\n"
        sys.stderr.write(out)
        sys.stderr.flush()
        self._lines_drawn = len(lines)

    def __enter__(self) -> "MultiBar":
        if not self._quiet:
            self._render_thread = threading.Thread(target=self._render_loop, daemon=True)
            self._render_thread.start()
        return self

    def wait(self) -> None:
        first_err: Optional[BaseException] = None
        for f in self._futures:
            try:
                f.result()
            except BaseException as e:  # keep first error, let the rest finish
                if first_err is None:
                    first_err = e
        if first_err is not None:
            raise first_err

    def __exit__(self, exc_type, exc, tb) -> None:
        self._pool.shutdown(wait=True)
        if self._render_thread is not None:
            self._stop_render.set()
            self._render_thread.join()
            self._render_once()
