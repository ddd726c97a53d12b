"""NornicDB-AMD command-line interface.

Parity: reference cmd/nornicdb/main.go cobra commands:
    serve | init | import | shell | decay  (main.go:71-210)

Usage:
    python -m nornicdb_amd serve [--data-dir DIR] [--config FILE]
    python -m nornicdb_amd init --data-dir DIR
    python -m nornicdb_amd import --data-dir DIR --file export.json
    python -m nornicdb_amd shell [--data-dir DIR]
    python -m nornicdb_amd decay --data-dir DIR
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import sys


def _open(args, cfg):
    from .db import open_db
    from .embed import create_embedder

    if getattr(cfg, "search_quant", ""):
        # SearchService reads the env at index construction
        os.environ.setdefault("NORNICDB_SEARCH_QUANT", cfg.search_quant)

    provider = "mock" if cfg.embedder == "mock" else cfg.embedder
    try:
        emb = create_embedder(provider, dims=cfg.embedding_dims)
    except Exception as e:
        print(f"embedder '{provider}' unavailable ({e}); using mock",
              file=sys.stderr)
        emb = create_embedder("mock", dims=cfg.embedding_dims)
    return open_db(args.data_dir or cfg.data_dir or None, embedder=emb,
                   dims=cfg.embedding_dims,
                   engine=getattr(cfg, "storage_engine", "disk") or "disk",
                   encryption_passphrase=cfg.encryption_passphrase)


def cmd_serve(args):
    from .bolt import BoltServer
    from .server import create_app
    from .auth import Authenticator
    from .utils import load_config
    import uvicorn

    cfg = load_config(args.config)
    cluster_node = None
    if getattr(args, "cluster_peers", None):
        # replicated mode (reference serve cluster flags + :7688 transport):
        # --cluster-id me --cluster-peers me=host:7688,n2=host2:7688,...
        from .db import DatabaseManager
        from .embed import create_embedder
        from .replication.cluster import ClusterNode
        from .storage import DiskEngine, MemoryEngine
        peers = {}
        for part in args.cluster_peers.split(","):
            nid, addr = part.split("=", 1)
            host, port = addr.rsplit(":", 1)
            peers[nid.strip()] = (host.strip(), int(port))
        me = args.cluster_id or sorted(peers)[0]
        data_dir = args.data_dir or cfg.data_dir
        base = DiskEngine(data_dir) if data_dir else MemoryEngine()
        cluster_node = ClusterNode(me, peers, base).start()
        try:
            emb = create_embedder(cfg.embedder, dims=cfg.embedding_dims)
        except Exception:
            emb = create_embedder("mock", dims=cfg.embedding_dims)
        mgr = DatabaseManager(cluster_node.replicated, embedder=emb,
                              dims=cfg.embedding_dims)
        print(f"cluster member {me} of {sorted(peers)} "
              f"(transport {peers[me][0]}:{peers[me][1]})")
    else:
        mgr = _open(args, cfg)
    auth = None
    if cfg.auth_enabled or args.auth:
        auth = Authenticator(mgr.get("system").engine)
        pw = auth.ensure_admin(password=cfg.initial_admin_password or None)
        if pw:
            print(f"initial admin user 'neo4j' password: {pw}")

    # TLS (reference pkg/security TLS middleware): --tls enables both
    # bolt+s and https with the configured or a generated self-signed pair
    ssl_ctx = None
    cert = key = None
    if getattr(args, "tls", False) or getattr(args, "tls_cert", None):
        from .utils.tls import ensure_self_signed, make_ssl_context
        cert = getattr(args, "tls_cert", None)
        key = getattr(args, "tls_key", None)
        if not (cert and key):
            cert, key = ensure_self_signed(
                args.data_dir or cfg.data_dir or ".")
            print(f"TLS: using self-signed pair {cert}")
        ssl_ctx = make_ssl_context(cert, key)

    app = create_app(mgr, auth=auth)
    bolt = BoltServer(lambda db: mgr.get(db).executor,
                      host=cfg.bolt_host, port=args.bolt_port or cfg.bolt_port,
                      authenticator=auth, ssl_context=ssl_ctx,
                      log_queries=(getattr(args, "log_queries", False)
                                   or cfg.log_queries),
                      tx_factory=lambda name: mgr.get(name).begin_tx())
    grpc_server = None
    if getattr(args, "grpc_port", None):
        from .server.nornic_grpc import serve as grpc_serve
        grpc_server, gport = grpc_serve(mgr, host=cfg.http_host,
                                        port=args.grpc_port)
        print(f"gRPC (NornicSearch) listening on {cfg.http_host}:{gport}")
    qdrant_grpc_server = None
    if getattr(args, "qdrant_grpc_port", None):
        # share the HTTP app's registry so REST + gRPC see one store
        from .server.qdrant_grpc import serve as qg_serve
        qdrant_grpc_server, qport, _ = qg_serve(
            app.state.qdrant, host=cfg.http_host,
            port=args.qdrant_grpc_port)
        print(f"gRPC (qdrant compat) listening on {cfg.http_host}:{qport}")

    async def main():
        await bolt.start()
        print(f"Bolt listening on {cfg.bolt_host}:{bolt.port}")
        http_port = args.http_port or cfg.http_port
        if os.environ.get("NORNICDB_HTTP_SERVER") == "uvicorn":
            config = uvicorn.Config(app, host=cfg.http_host, port=http_port,
                                    log_level="warning",
                                    ssl_certfile=cert, ssl_keyfile=key)
            server = uvicorn.Server(config)
            print(f"HTTP listening on {cfg.http_host}:{config.port}")
            await server.serve()
        else:
            # default: in-repo asyncio HTTP server (server/fasthttp.py) —
            # ~2x uvicorn/h11 throughput on the tx/GraphQL hot routes
            from .server.fasthttp import start_http_server
            sctx = None
            if cert and key:
                from .utils.tls import make_ssl_context
                sctx = make_ssl_context(cert, key)
            srv = await start_http_server(app, cfg.http_host, http_port,
                                          ssl_context=sctx)
            print(f"HTTP listening on {cfg.http_host}:{http_port}")
            async with srv:
                await srv.serve_forever()

    # graceful SIGTERM (systemd/docker stop, test harnesses): without
    # this the process dies mid-buffer and the finally never runs
    import signal

    def _term(_sig, _frm):
        raise KeyboardInterrupt

    signal.signal(signal.SIGTERM, _term)
    try:
        asyncio.run(main())
    except KeyboardInterrupt:
        pass
    finally:
        if grpc_server is not None:
            grpc_server.stop(0)
        if qdrant_grpc_server is not None:
            qdrant_grpc_server.stop(0)
        if cluster_node is not None:
            cluster_node.stop()
        mgr.close()


def cmd_init(args):
    from .utils import load_config
    cfg = load_config(args.config)
    mgr = _open(args, cfg)
    db = mgr.get()
    print(f"initialized database at {args.data_dir or '(memory)'}: "
          f"{db.engine.node_count()} nodes")
    mgr.close()


def cmd_import(args):
    from .utils import load_config
    from .storage import Edge, Node

    cfg = load_config(args.config)
    mgr = _open(args, cfg)
    eng = mgr.get(args.database).engine
    with open(args.file) as f:
        data = json.load(f)
    n_nodes = n_edges = 0
    for nd in data.get("nodes", []):
        try:
            eng.create_node(Node(id=str(nd.get("id", n_nodes)),
                                 labels=nd.get("labels", []),
                                 properties=nd.get("properties", {})))
            n_nodes += 1
        except Exception:
            pass
    for ed in data.get("relationships", data.get("edges", [])):
        try:
            eng.create_edge(Edge(id=str(ed.get("id", f"e{n_edges}")),
                                 type=ed.get("type", "RELATED"),
                                 start_node=str(ed.get("start", ed.get("startNode"))),
                                 end_node=str(ed.get("end", ed.get("endNode"))),
                                 properties=ed.get("properties", {})))
            n_edges += 1
        except Exception:
            pass
    print(f"imported {n_nodes} nodes, {n_edges} relationships")
    mgr.close()


def cmd_shell(args):
    from .utils import load_config
    cfg = load_config(args.config)
    mgr = _open(args, cfg)
    db = mgr.get(args.database)
    print("NornicDB-AMD shell — Cypher queries; :quit to exit")
    while True:
        try:
            line = input("nornicdb> ").strip()
        except (EOFError, KeyboardInterrupt):
            break
        if not line:
            continue
        if line in (":quit", ":exit", "quit", "exit"):
            break
        try:
            r = db.cypher(line)
            print("\t".join(r.columns))
            for row in r.rows[:100]:
                print("\t".join(str(v) for v in row))
            if len(r.rows) > 100:
                print(f"... {len(r.rows) - 100} more rows")
        except Exception as e:
            print(f"error: {e}")
    mgr.close()


def cmd_eval(args):
    """IR evaluation over JSON cases (reference cmd/eval)."""
    from .search.eval import EvalHarness
    from .utils import load_config
    cfg = load_config(args.config)
    mgr = _open(args, cfg)
    db = mgr.get(args.database)
    db.embed_queue.drain()
    harness = EvalHarness(lambda q, k: [r.id for r in
                                        db.search.search(query=q, k=k)])
    cases = EvalHarness.load_cases(args.cases)
    report = harness.run(cases)
    print(json.dumps(report, indent=2))
    mgr.close()


def cmd_train(args):
    """LoRA fine-tuning of the Heimdall model from DB content or JSONL
    (reference neural/train.py)."""
    import torch

    from .embed.tokenizer import HashTokenizer
    from .models.heimdall import HeimdallConfig, HeimdallModel
    from .neural import (InstructionDataset, LoRATrainer, TrainConfig,
                         export_merged, generate_dataset_from_db)
    from .utils import load_config

    cfg = load_config(args.config)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    mcfg = HeimdallConfig() if device == "cuda" else HeimdallConfig.tiny()
    model = HeimdallModel(mcfg).init_small()
    tok = HashTokenizer(mcfg.vocab_size, mcfg.max_position)
    if args.dataset:
        ds = InstructionDataset.from_jsonl(args.dataset, tok)
    else:
        mgr = _open(args, cfg)
        recs = generate_dataset_from_db(mgr.get(args.database))
        mgr.close()
        if not recs:
            print("no training data in database", file=sys.stderr)
            return
        ds = InstructionDataset(recs, tok)
    tr = LoRATrainer(model, TrainConfig(epochs=args.epochs,
                                        batch_size=args.batch_size,
                                        lr=args.lr), device=device)
    hist = tr.train(ds)
    if hist:
        print(f"trained {tr.step} steps; loss {hist[0]['loss']:.3f} -> "
              f"{hist[-1]['loss']:.3f}")
    merged = tr.merge()
    if args.out:
        export_merged(merged, args.out)
        print(f"exported merged model to {args.out}")


def cmd_decay(args):
    from .cognitive import DecayManager
    from .utils import load_config
    cfg = load_config(args.config)
    mgr = _open(args, cfg)
    dm = DecayManager(mgr.get(args.database).engine)
    stats = dm.run_cycle()
    print(json.dumps(stats))
    mgr.close()


def main(argv=None):
    p = argparse.ArgumentParser(prog="nornicdb-amd")
    p.add_argument("--config", default=None)
    sub = p.add_subparsers(dest="cmd", required=True)

    sp = sub.add_parser("serve")
    sp.add_argument("--data-dir", default=None)
    sp.add_argument("--bolt-port", type=int, default=None)
    sp.add_argument("--http-port", type=int, default=None)
    sp.add_argument("--grpc-port", type=int, default=None,
                    help="enable the native NornicSearch gRPC API")
    sp.add_argument("--qdrant-grpc-port", type=int, default=None,
                    help="enable the Qdrant-compatible gRPC endpoint "
                         "(Qdrant default: 6334)")
    sp.add_argument("--auth", action="store_true")
    sp.add_argument("--log-queries", action="store_true",
                    help="log every Cypher query with duration to stdout")
    sp.add_argument("--tls", action="store_true",
                    help="enable TLS for Bolt and HTTP (self-signed if no cert)")
    sp.add_argument("--tls-cert", default=None)
    sp.add_argument("--tls-key", default=None)
    sp.add_argument("--cluster-id", default=None,
                    help="this member's id in --cluster-peers")
    sp.add_argument("--cluster-peers", default=None,
                    help="id=host:port,... Raft cluster over TCP")
    sp.set_defaults(fn=cmd_serve)

    for name, fn in (("init", cmd_init), ("decay", cmd_decay)):
        sp = sub.add_parser(name)
        sp.add_argument("--data-dir", default=None)
        sp.add_argument("--database", default=None)
        sp.set_defaults(fn=fn)

    sp = sub.add_parser("import")
    sp.add_argument("--data-dir", default=None)
    sp.add_argument("--database", default=None)
    sp.add_argument("--file", required=True)
    sp.set_defaults(fn=cmd_import)

    sp = sub.add_parser("train")
    sp.add_argument("--data-dir", default=None)
    sp.add_argument("--database", default=None)
    sp.add_argument("--dataset", default=None, help="JSONL prompt/completion")
    sp.add_argument("--epochs", type=int, default=1)
    sp.add_argument("--batch-size", type=int, default=4)
    sp.add_argument("--lr", type=float, default=2e-4)
    sp.add_argument("--out", default=None, help="export dir for merged model")
    sp.set_defaults(fn=cmd_train)

    sp = sub.add_parser("eval")
    sp.add_argument("--data-dir", default=None)
    sp.add_argument("--database", default=None)
    sp.add_argument("--cases", required=True)
    sp.set_defaults(fn=cmd_eval)

    sp = sub.add_parser("shell")
    sp.add_argument("--data-dir", default=None)
    sp.add_argument("--database", default=None)
    sp.set_defaults(fn=cmd_shell)

    args = p.parse_args(argv)
    args.fn(args)


if __name__ == "__main__":
    main()
