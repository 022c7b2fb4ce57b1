"""FastAPI façade — the vantage6 REST surface over the in-process core.

Reference parity: SURVEY.md Appendix A.1/A.2 — ``POST /api/token/user``,
``/api/token/node``, ``/api/token/refresh``, task CRUD + kill, runs,
results, organizations, collaborations, nodes, users, studies, version,
health; bearer-JWT auth; pagination via ``page``/``per_page``.

The reference uses Flask + flask-restful + marshmallow; this image has
FastAPI + pydantic, speaking the same JSON shapes (SURVEY.md §8 stack note).
"""
from __future__ import annotations

import threading
import time
from typing import Optional

from fastapi import Depends, FastAPI, HTTPException, Request, Response, WebSocket
from pydantic import BaseModel, Field

from ..common import jwt as jwtlib
from .core import BadRequest, Identity, NotFound, ServerCore
from .permission import PermissionError_


# ------------------------------------------------------------ request bodies
class UserTokenBody(BaseModel):
    username: str
    password: str


class NodeTokenBody(BaseModel):
    api_key: str


class RecoverBody(BaseModel):
    username: str = ""
    token: str = ""
    password: str = ""


class RefreshBody(BaseModel):
    refresh_token: str


class ContainerTokenBody(BaseModel):
    task_id: int
    image: str


class TaskOrg(BaseModel):
    id: int
    input: str = ""


class TaskCreateBody(BaseModel):
    name: str = ""
    image: str
    collaboration_id: int
    description: str = ""
    databases: list[dict] = Field(default_factory=list)
    study_id: Optional[int] = None
    organizations: list[TaskOrg]
    method: str = ""


class OrgCreateBody(BaseModel):
    name: str
    domain: str = ""
    country: str = ""
    public_key: str = ""


class OrgPatchBody(BaseModel):
    public_key: Optional[str] = None
    name: Optional[str] = None
    domain: Optional[str] = None
    country: Optional[str] = None


class CollabCreateBody(BaseModel):
    name: str
    organization_ids: list[int]
    encrypted: bool = False


class NodeCreateBody(BaseModel):
    name: str
    collaboration_id: int
    organization_id: int
    device: str = "cpu"


class UserCreateBody(BaseModel):
    username: str
    password: str
    organization_id: int
    roles: list[str] = Field(default_factory=list)
    firstname: str = ""
    lastname: str = ""
    email: str = ""


class StudyCreateBody(BaseModel):
    name: str
    collaboration_id: int
    organization_ids: list[int]


class AlgorithmSubmitBody(BaseModel):
    name: str
    image: str
    description: str = ""


class AlgorithmReviewBody(BaseModel):
    approve: bool
    comment: str = ""


class RoleCreateBody(BaseModel):
    name: str
    rule_ids: list[int]
    organization_id: Optional[int] = None
    description: str = ""


class RunPatchBody(BaseModel):
    status: Optional[str] = None
    result: Optional[str] = None
    log: Optional[str] = None


def create_app(core: ServerCore, auth_rate_limit: int = 30,
               auth_rate_window: float = 60.0) -> FastAPI:
    app = FastAPI(title="vantage6_amd orchestrator", docs_url="/api/docs")

    # per-IP sliding-window rate limit on credential endpoints (reference:
    # flask-limiter on the server app) — brute-force protection upstream
    # of the per-account lockout. In-memory per process, like the
    # reference's default memory storage.
    _rl_hits: dict[str, list[float]] = {}
    _rl_lock = threading.Lock()

    @app.middleware("http")
    async def _auth_rate_limiter(request: Request, call_next):
        path = request.url.path
        if auth_rate_limit and (path.startswith("/api/token/")
                                or path.startswith("/api/recover/")):
            ip = request.client.host if request.client else "?"
            now = time.monotonic()
            with _rl_lock:
                hits = _rl_hits.setdefault(ip, [])
                while hits and now - hits[0] > auth_rate_window:
                    hits.pop(0)
                if len(hits) >= auth_rate_limit:
                    from starlette.responses import JSONResponse

                    return JSONResponse(
                        {"detail": "too many authentication attempts"},
                        status_code=429,
                        headers={"Retry-After": str(int(auth_rate_window))})
                hits.append(now)
        return await call_next(request)

    def ident(request: Request) -> Identity:
        auth = request.headers.get("Authorization", "")
        if not auth.startswith("Bearer "):
            raise HTTPException(401, "missing bearer token")
        try:
            return core.identity_from_token(auth[len("Bearer ") :])
        except jwtlib.JWTError as exc:
            raise HTTPException(401, str(exc))

    def wrap(fn, *args, **kwargs):
        try:
            return fn(*args, **kwargs)
        except PermissionError_ as exc:
            raise HTTPException(exc.status_code, str(exc))
        except NotFound as exc:
            raise HTTPException(404, str(exc))
        except BadRequest as exc:
            raise HTTPException(400, str(exc))
        except jwtlib.JWTError as exc:
            raise HTTPException(401, str(exc))

    # ------------------------------------------------------------- tokens
    @app.post("/api/token/user")
    def token_user(body: UserTokenBody):
        return wrap(core.authenticate_user, body.username, body.password)

    @app.post("/api/token/node")
    def token_node(body: NodeTokenBody):
        return wrap(core.authenticate_node, body.api_key)

    @app.post("/api/token/refresh")
    def token_refresh(body: RefreshBody):
        return wrap(core.refresh_token, body.refresh_token)

    @app.post("/api/token/container")
    def token_container(body: ContainerTokenBody, who: Identity = Depends(ident)):
        return wrap(core.token_for_container, who, body.task_id, body.image)

    # -------------------------------------------------------------- tasks
    @app.post("/api/task", status_code=201)
    def task_create(body: TaskCreateBody, who: Identity = Depends(ident)):
        return wrap(
            core.create_task, who, name=body.name, image=body.image,
            collaboration_id=body.collaboration_id, description=body.description,
            databases=body.databases, study_id=body.study_id,
            organizations=[o.model_dump() for o in body.organizations],
            method=body.method,
        )

    @app.get("/api/task/{task_id}")
    def task_get(task_id: int, who: Identity = Depends(ident)):
        return wrap(core.get_task, who, task_id)

    def paginate_headers(out, request: Request, response: Response):
        """RFC-5988 ``Link`` pagination headers (reference A.1) alongside
        the body's ``links`` block."""
        links = out.get("links") if isinstance(out, dict) else None
        if links:
            from urllib.parse import urlencode

            def u(p):
                q = dict(request.query_params)
                q["page"] = str(p)
                q["per_page"] = str(links["per_page"])
                return f"<{request.url.path}?{urlencode(q)}>"

            parts = [f'{u(1)}; rel="first"', f'{u(links["pages"])}; rel="last"']
            if links["page"] < links["pages"]:
                parts.insert(0, f'{u(links["page"] + 1)}; rel="next"')
            if links["page"] > 1:
                parts.insert(0, f'{u(links["page"] - 1)}; rel="prev"')
            response.headers["Link"] = ", ".join(parts)
            response.headers["Total-Count"] = str(links["total"])
        return out

    @app.get("/api/task")
    def task_list(request: Request, response: Response,
                  who: Identity = Depends(ident), page: int = 1,
                  per_page: int = 20, parent_id: Optional[int] = None,
                  job_id: Optional[int] = None,
                  collaboration_id: Optional[int] = None,
                  name: Optional[str] = None, status: Optional[str] = None):
        out = wrap(core.list_tasks, who, page=page, per_page=per_page,
                   parent_id=parent_id, job_id=job_id,
                   collaboration_id=collaboration_id, name=name,
                   status=status)
        return paginate_headers(out, request, response)

    @app.delete("/api/task/{task_id}")
    def task_kill(task_id: int, who: Identity = Depends(ident)):
        return wrap(core.kill_task, who, task_id)

    # --------------------------------------------------------------- runs
    @app.get("/api/run")
    def run_list(request: Request, response: Response,
                 who: Identity = Depends(ident), task_id: Optional[int] = None,
                 node_id: Optional[int] = None, state: Optional[str] = None,
                 page: int = 1, per_page: int = 50):
        out = wrap(core.list_runs, who, task_id=task_id, node_id=node_id,
                   state=state, page=page, per_page=per_page)
        return paginate_headers(out, request, response)

    @app.get("/api/run/{run_id}")
    def run_get(run_id: int, who: Identity = Depends(ident)):
        return wrap(core.get_run, who, run_id)

    @app.patch("/api/run/{run_id}")
    def run_patch(run_id: int, body: RunPatchBody, who: Identity = Depends(ident)):
        return wrap(core.patch_run, who, run_id,
                    **{k: v for k, v in body.model_dump().items() if v is not None})

    @app.get("/api/result")
    def result_list(request: Request, response: Response,
                    who: Identity = Depends(ident), task_id: int = 0,
                    page: int = 1, per_page: int = 50):
        out = wrap(core.list_results, who, task_id=task_id, page=page,
                   per_page=per_page)
        return paginate_headers(out, request, response)

    # ------------------------------------------------------------ entities
    @app.post("/api/organization", status_code=201)
    def org_create(body: OrgCreateBody, who: Identity = Depends(ident)):
        return wrap(core.create_organization, who, body.name, domain=body.domain,
                    country=body.country, public_key=body.public_key)

    @app.get("/api/organization")
    def org_list(request: Request, response: Response,
                 who: Identity = Depends(ident), page: int = 1,
                 per_page: int = 20):
        out = wrap(core.list_organizations, who, page=page, per_page=per_page)
        return paginate_headers(out, request, response)

    @app.get("/api/organization/{org_id}")
    def org_get(org_id: int, who: Identity = Depends(ident)):
        return wrap(core.get_organization, who, org_id)

    @app.patch("/api/organization/{org_id}")
    def org_patch(org_id: int, body: OrgPatchBody, who: Identity = Depends(ident)):
        out = None
        if body.public_key is not None:
            out = wrap(core.set_organization_public_key, who, org_id,
                       body.public_key)
        if (body.name is not None or body.domain is not None
                or body.country is not None):
            out = wrap(core.update_organization, who, org_id, name=body.name,
                       domain=body.domain, country=body.country)
        if out is None:
            raise HTTPException(400, "nothing to patch")
        return out

    @app.delete("/api/organization/{org_id}")
    def org_delete(org_id: int, who: Identity = Depends(ident),
                   delete_dependents: bool = False):
        return wrap(core.delete_organization, who, org_id,
                    delete_dependents=delete_dependents)

    @app.patch("/api/collaboration/{collab_id}")
    def collab_patch(collab_id: int, body: dict,
                     who: Identity = Depends(ident)):
        return wrap(core.update_collaboration, who, collab_id,
                    name=(body or {}).get("name"),
                    encrypted=(body or {}).get("encrypted"))

    @app.delete("/api/collaboration/{collab_id}")
    def collab_delete(collab_id: int, who: Identity = Depends(ident),
                      delete_dependents: bool = False):
        return wrap(core.delete_collaboration, who, collab_id,
                    delete_dependents=delete_dependents)

    @app.patch("/api/study/{study_id}")
    def study_patch(study_id: int, body: dict, who: Identity = Depends(ident)):
        return wrap(core.update_study, who, study_id,
                    name=(body or {}).get("name"),
                    organization_ids=(body or {}).get("organization_ids"))

    @app.delete("/api/study/{study_id}")
    def study_delete(study_id: int, who: Identity = Depends(ident)):
        return wrap(core.delete_study, who, study_id)

    @app.get("/api/role/{role_id}")
    def role_get(role_id: int, who: Identity = Depends(ident)):
        return wrap(core.get_role, who, role_id)

    @app.get("/api/rule/{rule_id}")
    def rule_get(rule_id: int, who: Identity = Depends(ident)):
        return wrap(core.get_rule, who, rule_id)

    @app.patch("/api/role/{role_id}")
    def role_patch(role_id: int, body: dict, who: Identity = Depends(ident)):
        return wrap(core.update_role, who, role_id,
                    name=(body or {}).get("name"),
                    description=(body or {}).get("description"),
                    rule_ids=(body or {}).get("rule_ids"))

    @app.delete("/api/role/{role_id}")
    def role_delete(role_id: int, who: Identity = Depends(ident)):
        return wrap(core.delete_role, who, role_id)

    @app.post("/api/collaboration", status_code=201)
    def collab_create(body: CollabCreateBody, who: Identity = Depends(ident)):
        return wrap(core.create_collaboration, who, body.name,
                    body.organization_ids, body.encrypted)

    @app.get("/api/collaboration")
    def collab_list(request: Request, response: Response,
                    who: Identity = Depends(ident), page: int = 1,
                    per_page: int = 20):
        out = wrap(core.list_collaborations, who, page=page, per_page=per_page)
        return paginate_headers(out, request, response)

    @app.post("/api/node", status_code=201)
    def node_create(body: NodeCreateBody, who: Identity = Depends(ident)):
        return wrap(core.create_node, who, body.name, body.collaboration_id,
                    body.organization_id, body.device)

    @app.get("/api/node")
    def node_list(request: Request, response: Response,
                  who: Identity = Depends(ident), page: int = 1,
                  per_page: int = 20,
                  collaboration_id: Optional[int] = None):
        out = wrap(core.list_nodes, who, page=page, per_page=per_page,
                   collaboration_id=collaboration_id)
        return paginate_headers(out, request, response)

    @app.get("/api/user")
    def user_list(request: Request, response: Response,
                  who: Identity = Depends(ident), page: int = 1,
                  per_page: int = 20, organization_id: Optional[int] = None):
        out = wrap(core.list_users, who, page=page, per_page=per_page,
                   organization_id=organization_id)
        return paginate_headers(out, request, response)

    @app.get("/api/user/{user_id}")
    def user_get(user_id: int, who: Identity = Depends(ident)):
        return wrap(core.get_user, who, user_id)

    @app.patch("/api/user/{user_id}")
    def user_patch(user_id: int, body: dict, who: Identity = Depends(ident)):
        allowed = {k: v for k, v in (body or {}).items() if k in (
            "password", "email", "firstname", "lastname", "roles")}
        return wrap(core.update_user, who, user_id, **allowed)

    @app.delete("/api/user/{user_id}")
    def user_delete(user_id: int, who: Identity = Depends(ident)):
        return wrap(core.delete_user, who, user_id)

    @app.patch("/api/node/{node_id}")
    def node_patch(node_id: int, body: dict, who: Identity = Depends(ident)):
        return wrap(core.update_node, who, node_id,
                    name=(body or {}).get("name"))

    @app.delete("/api/node/{node_id}")
    def node_delete(node_id: int, who: Identity = Depends(ident)):
        return wrap(core.delete_node, who, node_id)

    @app.post("/api/recover/node")
    def node_recover(body: dict, who: Identity = Depends(ident)):
        try:
            node_id = int(body["id"])
        except (KeyError, TypeError, ValueError):
            raise HTTPException(400, "body needs an integer 'id'")
        return wrap(core.reset_node_api_key, who, node_id)

    @app.get("/api/collaboration/{collab_id}")
    def collaboration_get(collab_id: int, who: Identity = Depends(ident)):
        return wrap(core.get_collaboration, who, collab_id)

    @app.post("/api/user", status_code=201)
    def user_create(body: UserCreateBody, who: Identity = Depends(ident)):
        return wrap(core.create_user, who, body.username, body.password,
                    body.organization_id, body.roles, firstname=body.firstname,
                    lastname=body.lastname, email=body.email)

    @app.get("/api/study")
    def study_list(request: Request, response: Response,
                   who: Identity = Depends(ident), page: int = 1,
                   per_page: int = 20,
                   collaboration_id: Optional[int] = None):
        out = wrap(core.list_studies, who, collaboration_id=collaboration_id,
                   page=page, per_page=per_page)
        return paginate_headers(out, request, response)

    @app.get("/api/study/{study_id}")
    def study_get(study_id: int, who: Identity = Depends(ident)):
        return wrap(core.get_study, who, study_id)

    @app.post("/api/study", status_code=201)
    def study_create(body: StudyCreateBody, who: Identity = Depends(ident)):
        return wrap(core.create_study, who, body.name, body.collaboration_id,
                    body.organization_ids)

    @app.post("/api/role", status_code=201)
    def role_create(body: "RoleCreateBody", who: Identity = Depends(ident)):
        return wrap(core.create_role, who, body.name, body.rule_ids,
                    body.organization_id, body.description)

    @app.get("/api/role")
    def role_list(request: Request, response: Response,
                  who: Identity = Depends(ident), page: int = 1,
                  per_page: int = 50):
        out = wrap(core.list_roles, who, page=page, per_page=per_page)
        return paginate_headers(out, request, response)

    @app.get("/api/rule")
    def rule_list(request: Request, response: Response,
                  who: Identity = Depends(ident), page: int = 1,
                  per_page: int = 200):
        out = wrap(core.list_rules, who, page=page, per_page=per_page)
        return paginate_headers(out, request, response)

    # ------------------------------------------------------ algorithm store
    @app.post("/api/algorithm", status_code=201)
    def algorithm_submit(body: AlgorithmSubmitBody, who: Identity = Depends(ident)):
        return wrap(core.store_submit_algorithm, who, body.name, body.image,
                    body.description)

    @app.get("/api/algorithm")
    def algorithm_list(who: Identity = Depends(ident), status: Optional[str] = None,
                       page: int = 1, per_page: int = 20):
        return wrap(core.store_list_algorithms, who, status=status, page=page,
                    per_page=per_page)

    @app.post("/api/algorithm/{algorithm_id}/review")
    def algorithm_review(algorithm_id: int, body: AlgorithmReviewBody,
                         who: Identity = Depends(ident)):
        return wrap(core.store_review_algorithm, who, algorithm_id,
                    body.approve, body.comment)

    # ------------------------------------------------------------ websocket
    # Reference parity: the Socket.IO '/tasks' namespace (SURVEY.md §3.2 L3,
    # A.4) — a client subscribes to its collaboration rooms and receives
    # new_task / algorithm_status_change / kill_containers / node-status
    # events pushed instead of polling. Rooms derive from the caller's
    # identity like the reference's on_connect.
    @app.websocket("/api/ws")
    async def events_ws(ws: WebSocket):
        import asyncio

        await ws.accept()
        try:
            token = ws.query_params.get("token") or (
                ws.headers.get("Authorization", "").removeprefix("Bearer ")
            )
            who = core.identity_from_token(token)
        except Exception:  # noqa: BLE001
            await ws.close(code=4401)
            return
        rooms = []
        if who.collaboration_id is not None:
            rooms.append(f"collaboration_{who.collaboration_id}")
            if who.organization_id is not None:
                rooms.append(
                    f"collaboration_{who.collaboration_id}"
                    f"_organization_{who.organization_id}"
                )
        elif who.type == "user":
            # users join every collaboration their org is in
            for c in core.list_collaborations(who, per_page=1000)["data"]:
                rooms.append(f"collaboration_{c['id']}")
        queue: "asyncio.Queue" = asyncio.Queue()
        loop = asyncio.get_event_loop()

        def forward(event, data, _room):
            loop.call_soon_threadsafe(queue.put_nowait,
                                      {"event": event, "data": data})

        subs = []
        for room in rooms:
            cb = (lambda e, d, r=room: forward(e, d, r))
            core.events.subscribe(room, cb)
            subs.append((room, cb))
        try:
            while True:
                msg = await queue.get()
                await ws.send_json(msg)
        except Exception:  # noqa: BLE001 — disconnects
            pass
        finally:
            for room, cb in subs:
                core.events.unsubscribe(room, cb)

    # ------------------------------------------------------------- recovery
    from . import mail as mail_mod

    mail_service = mail_mod.MailService()
    app.state.mail = mail_service  # tests read the outbox

    @app.post("/api/recover/lost")
    def recover_lost(body: RecoverBody):
        wrap(mail_mod.request_password_recovery, core, mail_service,
             body.username)
        return {"msg": "if the user exists, a recovery email was sent"}

    @app.post("/api/recover/reset")
    def recover_reset(body: RecoverBody):
        wrap(mail_mod.reset_password, core, body.token, body.password)
        return {"msg": "password reset"}

    @app.post("/api/recover/2fa/lost")
    def recover_2fa_lost(body: RecoverBody):
        wrap(mail_mod.request_mfa_reset, core, mail_service, body.username)
        return {"msg": "if the user exists, an mfa-reset email was sent"}

    @app.post("/api/recover/2fa/reset")
    def recover_2fa_reset(body: RecoverBody):
        return wrap(core.reset_mfa_with_token, body.token)

    @app.get("/api/column/{node_id}")
    def get_columns(node_id: int, label: str = "default",
                    who: Identity = Depends(ident)):
        return wrap(core.get_columns, who, node_id, label)

    # ----------------------------------------------------------------- ops
    @app.post("/api/ping")
    def ping(body: dict, who: Identity = Depends(ident)):
        body = body if isinstance(body, dict) else {}
        if body.get("offline"):
            wrap(core.node_set_offline, who.id)
        else:
            info = body.get("info")
            wrap(core.node_ping, who,
                 info if isinstance(info, dict) and info else None)
        return {"msg": "pong"}

    @app.get("/api/version")
    def version():
        return core.version()

    @app.get("/api/health")
    def health():
        return core.health()

    return app
