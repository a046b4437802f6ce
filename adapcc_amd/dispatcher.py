"""Config-artifact distribution (reference: dispatcher.py).

The reference scp'd XML/CSV artifacts between nodes. On a single MI355X
node every rank shares the filesystem, so distribution is a local copy; for
multi-node the same API shells out to scp like the reference. All runtime
artifact exchange (detect graphs, profiles, strategies) additionally flows
over torch.distributed object collectives, so the files are a convenience /
debugging surface, not a correctness dependency.
"""

from __future__ import annotations

import os
import shutil
import subprocess
from typing import Iterable, List


def _is_local(host: str) -> bool:
    return host in ("127.0.0.1", "localhost", os.uname().nodename)


class Dispatcher:
    def __init__(self, hosts: Iterable[str], workdir: str = ".") -> None:
        self.hosts: List[str] = list(dict.fromkeys(hosts))
        self.workdir = workdir

    def _send(self, path: str, host: str, dest: str) -> None:
        if _is_local(host):
            dest_path = os.path.join(self.workdir, dest)
            if os.path.abspath(path) != os.path.abspath(dest_path):
                os.makedirs(os.path.dirname(dest_path) or ".", exist_ok=True)
                shutil.copyfile(path, dest_path)
        else:
            subprocess.run(["scp", "-q", path, f"{host}:{dest}"], check=True)

    def dispatch_ip_table(self, path: str) -> None:
        for h in self.hosts:
            self._send(path, h, path)

    def dispatch_detected_topo(self, path: str) -> None:
        for h in self.hosts:
            self._send(path, h, path)

    def send_profiled_topo(self, path: str, root_host: str) -> None:
        self._send(path, root_host, path)

    def dispatch_strategy(self, path: str) -> None:
        for h in self.hosts:
            self._send(path, h, path)
