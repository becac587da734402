"""Local runtime: executes the control plane's Pods as real processes.

The reference's kind path delegates execution to a real kubelet + containerd
(install/kind/up.sh); this environment has neither, so the LocalRuntime plays
kubelet for the data plane: it watches Jobs/Deployments the controllers
create, maps the well-known first-party images to their Python entrypoints,
resolves the container-contract mounts (hostPath bucket volumes under the
kind bucket dir + params ConfigMaps), launches subprocesses with the PARAM_*
environment, and reflects completion/readiness back into Job/Pod/Deployment
status — making the reference's system test (test/system.sh: apply Model +
Server, wait ready, POST /v1/completions) runnable end-to-end with no
cluster.

Image → entrypoint map (our first-party replacements for the reference's
external images):
  *model-loader*   → substratus_amd.loader.model_loader
  *data-loader* / *dataset-loader* → substratus_amd.loader.dataset_loader
  *trainer*        → substratus_amd.train
  *server*         → substratus_amd.serve.server (uvicorn on a free port)
"""

from __future__ import annotations

import os
import socket
import subprocess
import sys
import threading
import time
from typing import Optional

import httpx

from ..kube.memapi import MemKube


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


class LocalRuntime:
    def __init__(self, kube: MemKube, bucket_dir: str):
        self.kube = kube
        self.bucket_dir = bucket_dir
        self.procs: dict[tuple, subprocess.Popen] = {}
        self.server_ports: dict[str, int] = {}
        self._jobs_running: set = set()
        self._lock = threading.Lock()

    # ------------------------------------------------------------- driving
    def sync(self) -> None:
        """One pass: start pending Jobs/Deployments/Pods, probe readiness,
        and reap processes whose backing object is gone (a kubelet would
        kill the container when its Pod is deleted — e.g. Notebook
        suspend)."""
        for key in list(self.procs):
            kind, ns, name = key
            if self.kube.try_get(kind, name, ns) is None:
                proc = self.procs.pop(key)
                self.server_ports.pop(f"{ns}/{name}", None)
                if proc.poll() is None:
                    proc.terminate()
                    try:
                        proc.wait(timeout=5)
                    except subprocess.TimeoutExpired:
                        proc.kill()
        for job in self.kube.list("Job"):
            self._sync_job(job)
        for dep in self.kube.list("Deployment"):
            self._sync_deployment(dep)
        for pod in self.kube.list("Pod"):
            self._sync_pod(pod)

    def run_until_quiet(self, mgr, timeout: float = 120.0,
                        poll: float = 0.2) -> None:
        """Drive manager + runtime together until both settle."""
        deadline = time.time() + timeout
        while time.time() < deadline:
            mgr.run_until_settled(timeout=5.0)
            self.sync()
            with self._lock:
                busy = bool(self._jobs_running)
            if not busy and not mgr.queue and not mgr._delayed:
                return
            time.sleep(poll)

    # ----------------------------------------------------------------- jobs
    def _sync_job(self, job: dict) -> None:
        name = job["metadata"]["name"]
        ns = job["metadata"].get("namespace", "default")
        key = ("Job", ns, name)
        complete = any(c.get("type") in ("Complete", "Failed") and
                       c.get("status") == "True"
                       for c in job.get("status", {}).get("conditions", []) or [])
        if complete:
            return
        with self._lock:
            if key in self._jobs_running:
                return
            self._jobs_running.add(key)
        threading.Thread(target=self._run_job, args=(job, key),
                         daemon=True).start()

    def _run_job(self, job: dict, key: tuple) -> None:
        name = job["metadata"]["name"]
        ns = job["metadata"].get("namespace", "default")
        try:
            pod_spec = job["spec"]["template"]["spec"]
            c = pod_spec["containers"][0]
            env, mounts = self._resolve(pod_spec, c, ns)
            env["JOB_NAME"] = name
            rc = self._exec_container(c, env, mounts)
            cond = {"type": "Complete" if rc == 0 else "Failed",
                    "status": "True"}
            self.kube.patch("Job", name, {"status": {"conditions": [cond]}},
                            ns)
        except Exception as e:  # noqa: BLE001
            try:
                self.kube.patch("Job", name, {"status": {"conditions": [
                    {"type": "Failed", "status": "True",
                     "message": str(e)}]}}, ns)
            except Exception:
                pass
        finally:
            with self._lock:
                self._jobs_running.discard(key)

    # ----------------------------------------------------------- deployments
    def _sync_deployment(self, dep: dict) -> None:
        name = dep["metadata"]["name"]
        ns = dep["metadata"].get("namespace", "default")
        key = ("Deployment", ns, name)
        if key in self.procs and self.procs[key].poll() is None:
            self._probe_deployment(dep, name, ns)
            return
        pod_spec = dep["spec"]["template"]["spec"]
        c = pod_spec["containers"][0]
        env, mounts = self._resolve(pod_spec, c, ns)
        port = _free_port()
        env["PORT"] = str(port)
        self.server_ports[f"{ns}/{name}"] = port
        proc = self._spawn_server(c, env, mounts, ns, name)
        self.procs[key] = proc
        # record the local endpoint on the Service for clients/tests
        svc = self.kube.try_get("Service", name, ns)
        if svc is not None:
            self.kube.patch("Service", name, {"metadata": {"annotations": {
                "substratus.ai/local-endpoint":
                    f"http://127.0.0.1:{port}"}}}, ns)

    # ------------------------------------------------------------------ pods
    def _sync_pod(self, pod: dict) -> None:
        """Long-running Pods (Notebooks: jupyter on 8888, readiness GET
        /api). Locally the notebook container runs the shim
        (serve/notebook_shim.py) — same contract surface."""
        name = pod["metadata"]["name"]
        ns = pod["metadata"].get("namespace", "default")
        c = pod["spec"]["containers"][0]
        ports = [p.get("containerPort") for p in c.get("ports", []) or []]
        if 8888 not in ports:
            return  # only notebook-shaped pods run locally
        key = ("Pod", ns, name)
        if key in self.procs and self.procs[key].poll() is None:
            self._probe_pod(pod, name, ns)
            return
        env, mounts = self._resolve(pod["spec"], c, ns)
        port = _free_port()
        env["PORT"] = str(port)
        env.setdefault("CONTENT_DIR",
                       mounts.get("/content/artifacts",
                                  self.bucket_dir))
        self.server_ports[f"{ns}/{name}"] = port
        lf = open(self.log_path(ns, name), "ab")
        self.procs[key] = subprocess.Popen(
            [sys.executable, "-m", "substratus_amd.serve.notebook_shim"],
            env=env, cwd=_repo_root(), stdout=lf, stderr=lf)
        self.kube.patch("Pod", name, {"metadata": {"annotations": {
            "substratus.ai/local-endpoint":
                f"http://127.0.0.1:{port}"}}}, ns)

    def _probe_pod(self, pod: dict, name: str, ns: str) -> None:
        port = self.server_ports.get(f"{ns}/{name}")
        if port is None:
            return
        try:
            ready = httpx.get(f"http://127.0.0.1:{port}/api",
                              timeout=2).status_code == 200
        except Exception:
            ready = False
        conds = pod.get("status", {}).get("conditions", []) or []
        cur = any(c.get("type") == "Ready" and c.get("status") == "True"
                  for c in conds)
        if ready != cur:
            self.kube.patch("Pod", name, {"status": {"conditions": [
                {"type": "Ready",
                 "status": "True" if ready else "False"}]}}, ns)

    def _probe_deployment(self, dep: dict, name: str, ns: str) -> None:
        port = self.server_ports.get(f"{ns}/{name}")
        if port is None:
            return
        try:
            r = httpx.get(f"http://127.0.0.1:{port}/", timeout=2)
            ready = r.status_code == 200
        except Exception:
            ready = False
        cur = (dep.get("status", {}).get("readyReplicas", 0) or 0)
        if ready and cur == 0:
            self.kube.patch("Deployment", name,
                            {"status": {"readyReplicas": 1}}, ns)
        elif not ready and cur:
            self.kube.patch("Deployment", name,
                            {"status": {"readyReplicas": 0}}, ns)

    # ------------------------------------------------------------ resolution
    def _resolve(self, pod_spec: dict, container: dict,
                 ns: str) -> tuple[dict, dict]:
        env = dict(os.environ)
        for e in container.get("env", []) or []:
            if "value" in e:
                env[e["name"]] = e["value"]
            elif "valueFrom" in e and "secretKeyRef" in e["valueFrom"]:
                ref = e["valueFrom"]["secretKeyRef"]
                sec = self.kube.try_get("Secret", ref["name"], ns)
                if sec:
                    env[e["name"]] = sec.get("stringData", {}).get(
                        ref["key"], "")
        mounts = {}
        vols = {v["name"]: v for v in pod_spec.get("volumes", []) or []}
        for m in container.get("volumeMounts", []) or []:
            v = vols.get(m["name"])
            if v is None:
                continue
            if "hostPath" in v:
                path = v["hostPath"]["path"]
                os.makedirs(path, exist_ok=True)
                mounts[m["mountPath"]] = path
            elif "configMap" in v and m.get("subPath") == "params.json":
                cm = self.kube.try_get("ConfigMap", v["configMap"]["name"], ns)
                if cm:
                    pdir = os.path.join(self.bucket_dir, "_params")
                    os.makedirs(pdir, exist_ok=True)
                    pfile = os.path.join(
                        pdir, f"{ns}-{v['configMap']['name']}.json")
                    with open(pfile, "w") as f:
                        f.write(cm["data"]["params.json"])
                    mounts["/content/params.json"] = pfile
        if "/content/params.json" in mounts:
            env["PARAMS_PATH"] = mounts["/content/params.json"]
        if "/content/artifacts" in mounts:
            env["ARTIFACTS_DIR"] = mounts["/content/artifacts"]
        if "/content/model" in mounts:
            env["MODEL_DIR"] = mounts["/content/model"]
        if "/content/data" in mounts:
            env["DATA_DIR"] = mounts["/content/data"]
        return env, mounts

    # ------------------------------------------------------------ execution
    def _entrypoint(self, image: str) -> Optional[list[str]]:
        img = image.lower()
        if "model-loader" in img:
            return [sys.executable, "-m", "substratus_amd.loader.model_loader"]
        if "data-loader" in img or "dataset-loader" in img:
            return [sys.executable, "-m",
                    "substratus_amd.loader.dataset_loader"]
        if "trainer" in img:
            return [sys.executable, "-m", "substratus_amd.train"]
        if "server" in img:
            return [sys.executable, "-m", "substratus_amd.serve.server"]
        return None

    @staticmethod
    def _gpu_count(container: dict) -> int:
        res = container.get("resources", {}) or {}
        lim = res.get("limits", {}) or {}
        try:
            return int(lim.get("amd.com/gpu", 0))
        except (TypeError, ValueError):
            return 0

    def _exec_container(self, container: dict, env: dict,
                        mounts: dict) -> int:
        cmd = self._entrypoint(container.get("image", ""))
        if cmd is None:
            # unknown image (e.g. kaniko builder): trivially succeed — there
            # is no container runtime to build real images locally
            return 0
        ngpu = self._gpu_count(container)
        if ngpu > 1 and cmd[1] == "-m":
            # multi-GPU Job: one rank per GPU over RCCL (container contract
            # "Multi-GPU (MI355X)" — the trainer entrypoint is DDP-aware)
            cmd = [sys.executable, "-m", "torch.distributed.run",
                   "--standalone", "--local-addr", "127.0.0.1",
                   f"--nproc-per-node={ngpu}", "-m", cmd[2]]
        args = []
        if "/content/artifacts" in mounts and "loader" in cmd[-1]:
            args = [mounts["/content/artifacts"]]
        proc = subprocess.run(cmd + args, env=env, cwd=_repo_root(),
                              capture_output=True, timeout=600)
        with open(self.log_path("default",
                                env.get("JOB_NAME", "job")), "ab") as lf:
            lf.write(proc.stdout)
            lf.write(proc.stderr)
        if proc.returncode != 0:
            sys.stderr.write(proc.stderr.decode()[-2000:])
        return proc.returncode

    def log_path(self, ns: str, name: str) -> str:
        d = os.path.join(self.bucket_dir, "_logs")
        os.makedirs(d, exist_ok=True)
        return os.path.join(d, f"{ns}-{name}.log")

    def _spawn_server(self, container: dict, env: dict,
                      mounts: dict, ns: str = "default",
                      name: str = "server") -> subprocess.Popen:
        cmd = self._entrypoint(container.get("image", "")) or [
            sys.executable, "-m", "substratus_amd.serve.server"]
        lf = open(self.log_path(ns, name), "ab")
        return subprocess.Popen(cmd, env=env, cwd=_repo_root(),
                                stdout=lf, stderr=lf)

    def stop(self) -> None:
        # snapshot: the platform's reconcile thread may still be spawning/
        # reaping concurrently (e.g. a Notebook suspend mid-shutdown) —
        # iterating the live dict races ("dictionary changed size")
        procs = list(self.procs.values())
        for proc in procs:
            if proc.poll() is None:
                proc.terminate()
        for proc in procs:
            try:
                proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                proc.kill()


def _repo_root() -> str:
    return os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
