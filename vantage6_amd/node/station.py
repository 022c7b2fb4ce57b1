"""Station daemon — one worker per data station (per GPU in production).

Reference parity: ``vantage6-node/vantage6/node/__init__.py::Node``
 — authenticate with api key, subscribe to task
events, sync open runs on boot (crash recovery), a listening worker thread
pops the task queue and executes runs, results are patched back with status
and captured logs; kill events cancel queued/active work.

MI355X mapping: instead of spawning a Docker container per run, the run's
algorithm package executes in-process on the station's pinned HIP device
(``torch.cuda.set_device``); the I/O file contract (INPUT_FILE/OUTPUT_FILE/
TOKEN_FILE + env map, Appendix A.3) is materialized under a per-job
temporary folder for algorithm-code portability.
"""
from __future__ import annotations

import json
import logging
import queue
import tempfile
import threading
import traceback
from pathlib import Path

from ..common import bytes_to_base64s
from ..common.config import StationConfig
from ..common.encryption import CryptorBase, RSACryptor
from ..common.serialization import deserialize, serialize
from ..common.task_status import RunStatus
from ..server.core import Identity, ServerCore
from ..algorithm.client import AlgorithmClient
from ..algorithm.exceptions import AlgorithmError, InputError
from ..algorithm.wrap import dispatch, resolve_module
from .proxy import NodeProxy
from ..utils.stdcapture import capture_thread_output

log = logging.getLogger(__name__)

_STOP = object()


class KilledInterrupt(BaseException):
    """Raised asynchronously inside an algorithm thread on kill_containers.

    BaseException so a broad ``except Exception`` inside user algorithm code
    cannot swallow the kill. The in-process analog of the reference's
    container kill (``DockerManager.kill_tasks``,
    ``vantage6-node/vantage6/node/docker/docker_manager.py``): the algorithm
    stops at its next Python bytecode boundary, the station thread and GPU
    are freed for the next task.
    """


def _async_raise(thread_ident: int, exc_type) -> None:
    """Inject ``exc_type`` into a running thread (CPython API)."""
    import ctypes

    res = ctypes.pythonapi.PyThreadState_SetAsyncExc(
        ctypes.c_ulong(thread_ident), ctypes.py_object(exc_type))
    if res > 1:  # "injected into more than one thread" — undo, per CPython docs
        ctypes.pythonapi.PyThreadState_SetAsyncExc(
            ctypes.c_ulong(thread_ident), None)


class Station:
    """In-process data-station worker."""

    def __init__(
        self,
        core: ServerCore,
        config: StationConfig,
        api_key: str,
        private_key_file: str | None = None,
        key_bits: int = 2048,
    ):
        self.core = core
        self.config = config
        self.api_key = api_key
        self.queue: "queue.Queue" = queue.Queue()
        self.identity: Identity | None = None
        self._thread: threading.Thread | None = None
        self._pool = None  # ThreadPoolExecutor: tasks run concurrently, like
        # the reference's one-Docker-container-per-run (a central task blocks
        # in wait_for_results while its partial runs on the same station)
        self._kill_flags: set[int] = set()  # run ids killed while queued/active
        self._active: dict[int, int] = {}  # run id -> executing thread ident
        self._lock = threading.Lock()
        self._inflight = 0
        if private_key_file:
            self.cryptor: CryptorBase = RSACryptor(private_key_file, bits=key_bits)
        else:
            self.cryptor = CryptorBase()
        # TEMPORARY_FOLDER root, shared across runs of the same job_id
        self._tmp_root = Path(tempfile.mkdtemp(prefix=f"v6amd_{config.name}_"))

    # ------------------------------------------------------------- lifecycle
    def start(self) -> None:
        token = self.core.authenticate_node(self.api_key)
        self.identity = self.core.identity_from_token(token["access_token"])
        # upload our public key so clients/nodes can encrypt for our org
        if isinstance(self.cryptor, RSACryptor):
            self.core.set_organization_public_key(
                self.identity, self.identity.organization_id,
                self.cryptor.public_key_str)
        cid, oid = self.identity.collaboration_id, self.identity.organization_id
        self.core.events.subscribe(f"collaboration_{cid}_organization_{oid}", self._on_event)
        self.core.events.subscribe(f"collaboration_{cid}", self._on_event)
        # crash recovery: re-queue open runs assigned to us
        for run in self.core.node_sync_open_runs(self.identity):
            self.queue.put(run["task_id"])
        from concurrent.futures import ThreadPoolExecutor

        self._pool = ThreadPoolExecutor(
            max_workers=8, thread_name_prefix=f"station-{self.config.name}-exec"
        )
        self._thread = threading.Thread(
            target=self._listening_worker, name=f"station-{self.config.name}", daemon=True
        )
        self._thread.start()
        self._stop_ping = threading.Event()
        self._ping_thread = threading.Thread(
            target=self._heartbeat, name=f"station-{self.config.name}-ping",
            daemon=True,
        )
        self._ping_thread.start()
        # immediate first node-info-update (columns etc.) — the heartbeat
        # loop's first tick is heartbeat_interval away
        try:
            self.core.node_ping(self.identity, {
                "device": self.config.device,
                "queue_depth": self.queue.qsize(),
                "inflight": 0,
                "columns": self._database_columns(),
            })
        except Exception:  # noqa: BLE001
            log.exception("initial node-info-update failed")

    heartbeat_interval: float = 15.0

    def _heartbeat(self) -> None:
        """Periodic last_seen ping + node-info-update (reference: socket
        ping loop + capability broadcast)."""
        while not self._stop_ping.wait(self.heartbeat_interval):
            try:
                with self._lock:
                    busy = self._inflight
                self.core.node_ping(self.identity, {
                    "device": self.config.device,
                    "queue_depth": self.queue.qsize(),
                    "inflight": busy,
                    "columns": self._database_columns(),
                })
            except Exception:  # noqa: BLE001
                log.exception("heartbeat failed")

    def _database_columns(self) -> dict:
        """Column names per database label (reference: /api/column — data
        discovery for the UI/researcher). Cached; only tabular sources
        report columns, tensor/synthetic report none."""
        if getattr(self, "_columns_cache", None) is not None:
            return self._columns_cache
        out: dict[str, list[str]] = {}
        for db in self.config.databases:
            try:
                if db.type in ("csv", "parquet", "excel", "sql"):
                    from ..algorithm.decorators import _load_database

                    df = _load_database(db.uri, db.type, {})
                    out[db.label] = [str(c) for c in df.columns]
            except Exception:  # noqa: BLE001 — discovery must not kill pings
                log.exception("column discovery failed for %s", db.label)
        self._columns_cache = out
        return out

    def stop(self, timeout: float = 10.0) -> None:
        if getattr(self, "_stop_ping", None) is not None:
            self._stop_ping.set()
        # preempt anything still executing (reference: daemon shutdown kills
        # its task containers); without this, non-daemon pool threads keep
        # the interpreter alive at exit
        with self._lock:
            active_tids = list(self._active.values())
            self._kill_flags.update(self._active.keys())
        for tid in active_tids:
            _async_raise(tid, KilledInterrupt)
        self.queue.put(_STOP)
        if self._thread:
            self._thread.join(timeout)
        if self._pool:
            self._pool.shutdown(wait=False)
        if self.identity:
            self.core.node_set_offline(self.identity.id)

    def join_idle(self, timeout: float = 60.0) -> bool:
        """Testing/ops helper: wait until the queue is drained."""
        import time

        t0 = time.monotonic()
        while time.monotonic() - t0 < timeout:
            with self._lock:
                busy = self._inflight
            if self.queue.empty() and busy == 0:
                return True
            time.sleep(0.01)
        return False

    # ---------------------------------------------------------------- events
    def _on_event(self, event: str, data: dict) -> None:
        if event == "new_task":
            if data.get("organization_id") == self.identity.organization_id:
                self.queue.put(data["task_id"])
        elif event == "kill_containers":
            to_interrupt = []
            with self._lock:
                for item in data.get("kill_list", []):
                    if item.get("organization_id") == self.identity.organization_id:
                        rid = item["run_id"]
                        self._kill_flags.add(rid)
                        tid = self._active.get(rid)
                        if tid is not None:
                            to_interrupt.append(tid)
            # preempt RUNNING algorithms (the reference kills the container)
            for tid in to_interrupt:
                _async_raise(tid, KilledInterrupt)

    # ------------------------------------------------------------- hot loop
    def _listening_worker(self) -> None:
        while True:
            item = self.queue.get()
            if item is _STOP:
                return
            with self._lock:
                self._inflight += 1
            self._pool.submit(self._run_one, item)

    def _run_one(self, task_id: int) -> None:
        try:
            self._execute_task(task_id)
        except Exception:  # noqa: BLE001 — worker must survive anything
            log.exception("station %s: task %s execution failed", self.config.name, task_id)
        finally:
            with self._lock:
                self._inflight -= 1

    def _my_run_for_task(self, task_id: int) -> dict | None:
        runs = self.core.list_runs(self.identity, task_id=task_id)["data"]
        for r in runs:
            if r["node_id"] == self.identity.id:
                return r
        return None

    def _execute_task(self, task_id: int) -> None:
        run = self._my_run_for_task(task_id)
        if run is None:
            return
        run_id = run["id"]
        # kill semantics: a queued run is dropped here; a RUNNING algorithm
        # is pre-empted via an async KilledInterrupt injected into its
        # executor thread (see _on_event) — the in-process analog of the
        # reference killing the Docker container. The server marks the run
        # killed (terminal) and any late patch from us cannot resurrect it.
        with self._lock:
            if run_id in self._kill_flags:
                self._kill_flags.discard(run_id)
                return
        if run["status"] != RunStatus.PENDING.value:
            return  # already handled (e.g. killed while queued)
        task = self.core.get_task(self.identity, task_id)
        self.core.patch_run(self.identity, run_id, status=RunStatus.INITIALIZING.value)

        # -- policy checks (reference: DockerManager allowed-images policy +
        #    allowed-algorithm-stores)
        allowed = self.config.policies.get("allowed_algorithms")
        if allowed is not None and task["image"] not in allowed:
            self.core.patch_run(
                self.identity, run_id, status=RunStatus.NOT_ALLOWED.value,
                log=f"algorithm {task['image']!r} not in station allow-list",
            )
            return
        if self.config.policies.get("require_approved_algorithms") and \
                not self.core.algorithm_is_approved(task["image"]):
            self.core.patch_run(
                self.identity, run_id, status=RunStatus.NOT_ALLOWED.value,
                log=f"algorithm {task['image']!r} is not store-approved",
            )
            return
        # reference 4.x node policies: restrict WHO may run here, not just
        # what — allowed_users / allowed_organizations are id allow-lists
        # against the task initiator
        allowed_users = self.config.policies.get("allowed_users")
        if allowed_users is not None and \
                task.get("init_user_id") not in allowed_users:
            self.core.patch_run(
                self.identity, run_id, status=RunStatus.NOT_ALLOWED.value,
                log=f"user {task.get('init_user_id')} not in station "
                    f"allowed_users policy",
            )
            return
        allowed_orgs = self.config.policies.get("allowed_organizations")
        if allowed_orgs is not None and \
                task.get("init_org_id") not in allowed_orgs:
            self.core.patch_run(
                self.identity, run_id, status=RunStatus.NOT_ALLOWED.value,
                log=f"organization {task.get('init_org_id')} not in station "
                    f"allowed_organizations policy",
            )
            return
        try:
            module = resolve_module(task["image"])
        except InputError as exc:
            self.core.patch_run(
                self.identity, run_id, status=RunStatus.NO_IMAGE.value, log=str(exc)
            )
            return

        # -- decrypt input (encrypted to this org's public key)
        try:
            payload = deserialize(self.cryptor.decrypt_str_to_bytes(run["input"]))
        except Exception as exc:  # noqa: BLE001
            self.core.patch_run(
                self.identity, run_id, status=RunStatus.START_FAILED.value,
                log=f"input decryption/deserialization failed: {exc}",
            )
            return

        # -- build runtime contract (Appendix A.3)
        token = self.core.token_for_container(self.identity, task_id, task["image"])
        container_ident = self.core.identity_from_token(token["container_token"])
        job_dir = self._tmp_root / f"job_{task['job_id']}"
        job_dir.mkdir(parents=True, exist_ok=True)
        run_dir = job_dir / f"run_{run_id}"
        run_dir.mkdir(exist_ok=True)
        input_file = run_dir / "input"
        output_file = run_dir / "output"
        token_file = run_dir / "token"
        input_file.write_bytes(serialize(payload))
        token_file.write_text(token["container_token"])
        env = {
            "INPUT_FILE": str(input_file),
            "OUTPUT_FILE": str(output_file),
            "TOKEN_FILE": str(token_file),
            "TEMPORARY_FOLDER": str(job_dir),
            "TASK_ID": str(task_id),
            "NODE_ID": str(self.identity.id),
            "ORGANIZATION_ID": str(self.identity.organization_id),
            "COLLABORATION_ID": str(self.identity.collaboration_id),
            "PKG_NAME": module.__name__,
            "HOST": "in-process", "PORT": "0", "API_PATH": "/api",
        }
        requested = [d.get("label", "default") for d in task.get("databases", [])]
        if requested:
            env["USER_REQUESTED_DATABASE_LABELS"] = ",".join(requested)
        for d in task.get("databases", []):
            if d.get("preprocessing"):
                env[f"{d.get('label', 'default').upper()}_PREPROCESSING"] = \
                    json.dumps(d["preprocessing"])
        for db in self.config.databases:
            env[f"{db.label.upper()}_DATABASE_URI"] = db.uri
            env[f"{db.label.upper()}_DATABASE_TYPE"] = db.type
            if db.label == "default":
                env["DATABASE_URI"] = db.uri
                env["DATABASE_TYPE"] = db.type

        collab_encrypted = isinstance(self.cryptor, RSACryptor)
        proxy = NodeProxy(self.core, container_ident, self.cryptor,
                          collab_encrypted, task["image"])
        client = AlgorithmClient(proxy)

        # -- execute on the pinned device, capturing stdout/stderr as the log
        self.core.patch_run(self.identity, run_id, status=RunStatus.ACTIVE.value)
        with self._lock:
            if run_id in self._kill_flags:  # killed between queue-pop and start
                self._kill_flags.discard(run_id)
                return
            self._active[run_id] = threading.get_ident()
        import io

        buf = io.StringIO()  # in case the kill lands before capture starts
        try:
            with capture_thread_output() as buf:
                self._bind_device()
                result = dispatch(module, payload, env, client)
            blob = serialize(result)
            output_file.write_bytes(blob)
            # encrypt result for the task's initiating organization
            result_str = self._encrypt_result(blob, task["init_org_id"])
            self.core.patch_run(
                self.identity, run_id, status=RunStatus.COMPLETED.value,
                result=result_str, log=buf.getvalue(),
            )
        except KilledInterrupt:
            # status already KILLED server-side; log the preemption (the
            # terminal-killed guard keeps the status from changing)
            try:
                self.core.patch_run(
                    self.identity, run_id, status=RunStatus.KILLED.value,
                    log=buf.getvalue() + "\nalgorithm pre-empted by kill request",
                )
            except Exception:  # noqa: BLE001
                log.exception("post-kill patch failed for run %s", run_id)
        except AlgorithmError as exc:
            self.core.patch_run(
                self.identity, run_id, status=RunStatus.FAILED.value,
                log=buf.getvalue() + f"\n{type(exc).__name__}: {exc}",
            )
        except Exception:  # noqa: BLE001
            self.core.patch_run(
                self.identity, run_id, status=RunStatus.CRASHED.value,
                log=buf.getvalue() + "\n" + traceback.format_exc(),
            )
        finally:
            with self._lock:
                self._active.pop(run_id, None)
                self._kill_flags.discard(run_id)

    def _encrypt_result(self, blob: bytes, init_org_id: int | None) -> str:
        """Encrypt the result for the task's initiating organization.

        In an encrypted collaboration a missing recipient key is an ERROR
        (fail the run), never a silent downgrade to plaintext — the same
        contract NodeProxy._encrypt_for_org enforces on the input path.
        """
        if isinstance(self.cryptor, RSACryptor):
            if init_org_id is None:
                raise RuntimeError(
                    "encrypted collaboration but task has no initiating "
                    "organization to encrypt the result for")
            pubkey = self.core.organization_public_key(init_org_id)
            if not pubkey:
                raise RuntimeError(
                    f"encrypted collaboration but organization {init_org_id} "
                    "has no public key uploaded — refusing to send the "
                    "result in plaintext")
            return self.cryptor.encrypt_bytes_to_str(blob, pubkey)
        return bytes_to_base64s(blob)

    def _bind_device(self) -> None:
        dev = self.config.device
        if dev == "auto":
            try:
                import torch

                dev = "cuda:0" if torch.cuda.is_available() else "cpu"
            except ImportError:
                dev = "cpu"
        if dev.startswith("cuda"):
            import torch

            torch.cuda.set_device(dev)
