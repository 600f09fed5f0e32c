"""HTTP frontend (ref: kolibrie-http-server/src/main.rs, 1 345 LoC).

Endpoints (parity with main.rs:594-637):
  GET  /                  — playground UI (:598)
  POST /query             — SPARQL query, JSON results; updates rejected (:602)
  POST /update            — SPARQL update
  POST /rsp-query         — stateless RSP-QL evaluation (:609)
  POST /rsp/register      — create an RSP session from an RSP-QL query (:616)
  POST /rsp/push          — push events into a session's streams (:623)
  GET  /rsp/events/{id}   — SSE stream of window results (:829)

The reference hand-rolls HTTP/1.1 over TcpStream with thread-per-connection;
here the host runtime uses FastAPI/uvicorn (the MI355X build's C++ lives in
the compute path, not the socket loop).  CORS is open like the reference.
"""
# (no `from __future__ import annotations`: FastAPI must resolve the
# `Request` annotation eagerly inside the create_app closure)

import asyncio
import json
import logging
import os
import threading
import uuid
from typing import Dict, List, Optional

from ..storage.database import SparqlDatabase

_log = logging.getLogger("kolibrie_amd.http")

MAX_BODY = 16 * 1024 * 1024  # request size cap (ref main.rs:432)


def sparql_json_results(var_names: List[str], rows: List[List[str]]) -> dict:
    """SPARQL 1.1 JSON results."""
    bindings = []
    for r in rows:
        b = {}
        for v, val in zip(var_names, r):
            kind = "uri" if val.startswith(("http://", "https://", "urn:")) \
                else ("bnode" if val.startswith("_:") else "literal")
            b[v] = {"type": kind, "value": val}
        bindings.append(b)
    return {"head": {"vars": var_names}, "results": {"bindings": bindings}}


class _Session:
    def __init__(self, engine):
        self.engine = engine
        self.events: "asyncio.Queue[str]" = None  # set lazily per loop
        self.buffer: List[str] = []
        self.lock = threading.Lock()

    def emit(self, rows):
        payload = json.dumps({"rows": [list(r) for r in rows]})
        with self.lock:
            self.buffer.append(payload)


def create_app(db: Optional[SparqlDatabase] = None, device: str = "cpu"):
    from fastapi import FastAPI, HTTPException, Request
    from fastapi.middleware.cors import CORSMiddleware
    from fastapi.responses import HTMLResponse, JSONResponse, StreamingResponse

    app = FastAPI(title="kolibrie_amd")
    app.add_middleware(
        CORSMiddleware, allow_origins=["*"], allow_methods=["*"],
        allow_headers=["*"],
    )
    state_db = db if db is not None else SparqlDatabase(device=device)
    sessions: Dict[str, _Session] = {}
    db_lock = threading.Lock()

    async def _body_text(request: Request) -> str:
        body = await request.body()
        if len(body) > MAX_BODY:
            raise HTTPException(413, "request body too large")
        try:
            data = json.loads(body)
            if isinstance(data, dict) and "query" in data:
                return data["query"]
        except json.JSONDecodeError:
            pass
        text = body.decode("utf-8", "replace")
        ctype = request.headers.get("content-type", "")
        if "application/x-www-form-urlencoded" in ctype:
            # SPARQL-protocol form encoding: query=...&... (ref
            # http_sparql_query_encodings_use_the_unified_query_executor)
            from urllib.parse import parse_qs
            form = parse_qs(text)
            if "query" in form:
                return form["query"][0]
            if "update" in form:
                return form["update"][0]
        return text

    @app.get("/", response_class=HTMLResponse)
    async def index():
        path = os.path.join(os.path.dirname(__file__), "playground.html")
        with open(path, "r", encoding="utf-8") as f:
            return f.read()

    @app.post("/query")
    async def query(request: Request):
        sparql = await _body_text(request)
        from ..engine.query import execute_sparql_query
        from ..parsing.sparql import ParseError, parse_combined_query
        try:
            with db_lock:
                rows = execute_sparql_query(sparql, state_db)
            cq = parse_combined_query(sparql)
            names = [p.output_name() for p in cq.select.variables] \
                if cq.select and cq.select.variables and not cq.select.select_star \
                else [f"v{i}" for i in range(len(rows[0]))] if rows else []
            return JSONResponse(sparql_json_results(names, rows))
        except ValueError as e:
            _log.warning("query rejected: %s", e)
            raise HTTPException(400, str(e))

    @app.post("/update")
    async def update(request: Request):
        sparql = await _body_text(request)
        try:
            with db_lock:
                state_db.query(sparql)
            return JSONResponse({"status": "ok"})
        except ValueError as e:
            _log.warning("update rejected: %s", e)
            raise HTTPException(400, str(e))

    @app.post("/rsp-query")
    async def rsp_query(request: Request):
        """Stateless: build an engine, replay supplied events, return
        results (ref :609)."""
        body = json.loads(await request.body())
        from ..rsp.builder import RSPBuilder
        results: List = []
        eng = (RSPBuilder(device=device)
               .add_rsp_ql_query(body["query"])
               .add_consumer(lambda rows: results.append(rows))
               .build())
        for ev in body.get("events", []):
            eng.add_to_stream(ev["stream"],
                              (ev["s"], ev["p"], ev["o"]), int(ev["ts"]))
        eng.flush_windows()
        return JSONResponse({"results": [[list(r) for r in rs]
                                         for rs in results]})

    @app.post("/rsp/register")
    async def rsp_register(request: Request):
        body = json.loads(await request.body())
        from ..rsp.builder import RSPBuilder
        sid = uuid.uuid4().hex[:12]
        builder = RSPBuilder(device=device).add_rsp_ql_query(body["query"])
        for rule in body.get("rules", []):
            builder.add_sparql_rules(rule)
        if body.get("static"):
            builder.add_static_ntriples(body["static"])
        sess = _Session(None)
        builder.add_consumer(sess.emit)
        sess.engine = builder.build()
        sessions[sid] = sess
        return JSONResponse({"session": sid})

    @app.post("/rsp/push")
    async def rsp_push(request: Request):
        body = json.loads(await request.body())
        sess = sessions.get(body.get("session", ""))
        if sess is None:
            raise HTTPException(404, "unknown session")
        for ev in body.get("events", []):
            sess.engine.add_to_stream(
                ev["stream"], (ev["s"], ev["p"], ev["o"]), int(ev["ts"]))
        return JSONResponse({"status": "ok"})

    @app.get("/rsp/events/{sid}")
    async def rsp_events(sid: str):
        sess = sessions.get(sid)
        if sess is None:
            raise HTTPException(404, "unknown session")

        async def stream():
            idx = 0
            while True:
                with sess.lock:
                    pending = sess.buffer[idx:]
                    idx = len(sess.buffer)
                for payload in pending:
                    yield f"data: {payload}\n\n"
                await asyncio.sleep(0.05)

        return StreamingResponse(stream(), media_type="text/event-stream")

    app.state.db = state_db
    app.state.sessions = sessions
    return app


def main():
    import argparse
    import uvicorn
    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8080)
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--load", default=None, help="RDF file to pre-load")
    args = ap.parse_args()
    db = SparqlDatabase(device=args.device)
    if args.load:
        db.load_file(args.load)
    uvicorn.run(create_app(db, args.device), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
