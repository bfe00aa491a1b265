"""Agent package manager + process lifecycle (reference parity: C29/C30 —
`af install/add/run/stop/logs/uninstall`, installed.json registry, venv-aware
process spawning with free-port allocation and readiness wait)."""
from __future__ import annotations

import json
import os
import shutil
import signal
import socket
import subprocess
import sys
import tarfile
import time
import zipfile
from pathlib import Path

import yaml


class PackageRegistry:
    def __init__(self, data_dir: str):
        self.root = Path(data_dir)
        self.pkg_dir = self.root / "packages"
        self.pkg_dir.mkdir(parents=True, exist_ok=True)
        self.index_path = self.root / "installed.json"

    def _load(self) -> dict:
        if self.index_path.exists():
            return json.loads(self.index_path.read_text())
        return {}

    def _save(self, idx: dict) -> None:
        self.index_path.write_text(json.dumps(idx, indent=2))

    @staticmethod
    def _read_metadata(path: Path) -> dict:
        meta = {}
        yml = path / "agentfield.yaml"
        if yml.exists():
            meta = yaml.safe_load(yml.read_text()) or {}
        meta.setdefault("name", path.name)
        meta.setdefault("entrypoint", "agent.py")
        return meta

    def install(self, source: str, name: str | None = None) -> dict:
        """Install from a local directory, archive (.zip/.tar.gz) or git URL
        (git clone — works for local/file:// repos offline)."""
        src = Path(source)
        if src.is_dir() and not (src / ".git").exists():
            staged = src
        elif src.is_file() and src.suffix == ".zip":
            staged = self.pkg_dir / (name or src.stem)
            with zipfile.ZipFile(src) as z:
                root = staged.resolve()
                for m in z.namelist():
                    # zip-slip guard: every member must resolve inside the
                    # target directory (reject ../ traversal and abs paths)
                    if not (root / m).resolve().is_relative_to(root):
                        raise ValueError(f"unsafe archive member: {m}")
                z.extractall(staged)
        elif src.is_file() and (src.name.endswith(".tar.gz")
                                or src.suffix == ".tgz"):
            staged = self.pkg_dir / (name or src.name.split(".")[0])
            with tarfile.open(src) as t:
                # 'data' filter rejects traversal, absolute names, devices
                # and out-of-tree links (tar-slip)
                t.extractall(staged, filter="data")
        else:  # treat as git URL / repo path
            target = self.pkg_dir / (name or Path(source).stem)
            if target.exists():
                shutil.rmtree(target)
            subprocess.run(["git", "clone", "--depth", "1", source,
                            str(target)], check=True, capture_output=True)
            staged = target
        meta = self._read_metadata(staged)
        pkg_name = name or meta["name"]
        dest = self.pkg_dir / pkg_name
        if staged != dest:
            if dest.exists():
                shutil.rmtree(dest)
            shutil.copytree(staged, dest, dirs_exist_ok=True)
        idx = self._load()
        idx[pkg_name] = {
            "name": pkg_name,
            "path": str(dest),
            "entrypoint": meta["entrypoint"],
            "source": str(source),
            "installed_at": time.time(),
        }
        self._save(idx)
        return idx[pkg_name]

    def uninstall(self, name: str) -> bool:
        idx = self._load()
        ent = idx.pop(name, None)
        if ent is None:
            return False
        self._save(idx)
        p = Path(ent["path"])
        if p.exists() and self.pkg_dir in p.parents:
            shutil.rmtree(p)
        return True

    def get(self, name: str) -> dict | None:
        return self._load().get(name)

    def list(self) -> list[dict]:
        return list(self._load().values())


class PortManager:
    @staticmethod
    def free_port(host: str = "127.0.0.1") -> int:
        s = socket.socket()
        s.bind((host, 0))
        port = s.getsockname()[1]
        s.close()
        return port


class ProcessManager:
    """Runs installed agent packages as OS processes with log capture,
    readiness wait, and pid tracking that survives CLI restarts."""

    def __init__(self, data_dir: str):
        self.root = Path(data_dir)
        self.run_dir = self.root / "run"
        self.log_dir = self.root / "logs"
        self.run_dir.mkdir(parents=True, exist_ok=True)
        self.log_dir.mkdir(parents=True, exist_ok=True)

    def _pidfile(self, name: str) -> Path:
        return self.run_dir / f"{name}.json"

    @staticmethod
    def _python_for(pkg_path: Path) -> str:
        for venv in (".venv", "venv"):
            cand = pkg_path / venv / "bin" / "python"
            if cand.exists():
                return str(cand)
        return sys.executable

    def start(self, pkg: dict, agentfield_url: str,
              port: int | None = None, extra_env: dict | None = None) -> dict:
        name = pkg["name"]
        if self.status(name).get("running"):
            return self.status(name)
        port = port or PortManager.free_port()
        path = Path(pkg["path"])
        log = self.log_dir / f"{name}.log"
        import agentfield_amd
        repo_root = str(Path(agentfield_amd.__file__).resolve().parent.parent)
        pypath = repo_root + os.pathsep + os.environ.get("PYTHONPATH", "")
        env = {**os.environ, "AGENTFIELD_URL": agentfield_url,
               "AGENT_PORT": str(port), "PYTHONPATH": pypath,
               **(extra_env or {})}
        runner = (
            "import importlib.util, os, sys\n"
            f"spec = importlib.util.spec_from_file_location('af_pkg', r'{path / pkg['entrypoint']}')\n"
            "m = importlib.util.module_from_spec(spec); spec.loader.exec_module(m)\n"
            "from agentfield_amd.sdk import Agent\n"
            "a = [v for v in vars(m).values() if isinstance(v, Agent)][0]\n"
            "a.serve(port=int(os.environ['AGENT_PORT']))\n")
        with open(log, "ab") as lf:
            proc = subprocess.Popen(
                [self._python_for(path), "-c", runner],
                cwd=str(path), env=env, stdout=lf, stderr=lf,
                start_new_session=True)
        info = {"name": name, "pid": proc.pid, "port": port,
                "base_url": f"http://127.0.0.1:{port}",
                "log": str(log), "started_at": time.time()}
        self._pidfile(name).write_text(json.dumps(info))
        return info

    def wait_ready(self, name: str, timeout: float = 30.0) -> bool:
        import httpx
        info = self.status(name)
        if not info.get("running"):
            return False
        deadline = time.time() + timeout
        url = info["base_url"] + "/health"
        while time.time() < deadline:
            try:
                if httpx.get(url, timeout=1.0).status_code == 200:
                    return True
            except httpx.HTTPError:
                pass
            if not self._alive(info["pid"]):
                return False
            time.sleep(0.1)
        return False

    @staticmethod
    def _alive(pid: int) -> bool:
        try:
            os.kill(pid, 0)
            return True
        except OSError:
            return False

    def status(self, name: str) -> dict:
        pf = self._pidfile(name)
        if not pf.exists():
            return {"name": name, "running": False}
        info = json.loads(pf.read_text())
        info["running"] = self._alive(info["pid"])
        return info

    def stop(self, name: str, timeout: float = 5.0) -> bool:
        info = self.status(name)
        if not info.get("running"):
            self._pidfile(name).unlink(missing_ok=True)
            return False
        os.kill(info["pid"], signal.SIGTERM)
        deadline = time.time() + timeout
        while time.time() < deadline and self._alive(info["pid"]):
            time.sleep(0.05)
        if self._alive(info["pid"]):
            os.kill(info["pid"], signal.SIGKILL)
        self._pidfile(name).unlink(missing_ok=True)
        return True

    def logs(self, name: str, lines: int = 50) -> str:
        log = self.log_dir / f"{name}.log"
        if not log.exists():
            return ""
        content = log.read_text(errors="replace").splitlines()
        return "\n".join(content[-lines:])
