"""creditcore CLI — the local train -> package -> serve lifecycle.

Replaces the reference's Databricks bundle + GitHub Actions + Bicep pipeline
 with one-command entrypoints on a single 8x MI355X node:

    python -m creditcore train  [--model-dir ./model] [--max-evals 10] ...
    python -m creditcore pack   --model-dir ./model --out ./model/packed.npz
    python -m creditcore serve  [--model-directory ./model] [--port 5000] ...
    python -m creditcore smoke  [--url http://127.0.0.1:5000]
"""

from __future__ import annotations

import argparse
import json
import sys


def _cmd_train(argv):
    p = argparse.ArgumentParser(prog="creditcore train")
    p.add_argument("--model-dir", default="./model")
    p.add_argument("--model-name", default="credit-default-uci-custom")
    p.add_argument("--registry-root", default=None)
    p.add_argument("--max-evals", type=int, default=10)
    p.add_argument("--n-rows", type=int, default=20_000)
    p.add_argument("--seed", type=int, default=2024)
    p.add_argument("--no-register", action="store_true")
    a = p.parse_args(argv)
    from .train import train_and_register

    uri = train_and_register(
        model_dir=a.model_dir,
        model_name=a.model_name,
        registry_root=a.registry_root,
        max_evals=a.max_evals,
        n_rows=a.n_rows,
        seed=a.seed,
        register=not a.no_register,
    )
    print(uri)


def _cmd_pack(argv):
    p = argparse.ArgumentParser(prog="creditcore pack")
    p.add_argument("--model-dir", default="./model")
    p.add_argument("--out", default=None)
    a = p.parse_args(argv)
    from .pack import pack_pyfunc_dir

    packed = pack_pyfunc_dir(a.model_dir)
    out = a.out or f"{a.model_dir.rstrip('/')}/packed.npz"
    packed.save(out)
    print(
        json.dumps(
            {
                "out": out,
                "cls_trees": packed.cls_n_trees,
                "cls_nodes": int(len(packed.cls_nodes)),
                "if_trees": packed.if_n_trees,
                "n_onehot": packed.n_onehot,
                "n_ref": packed.n_ref,
            }
        )
    )


def _cmd_serve(argv):
    from .serve import main as serve_main

    serve_main(argv)


def _cmd_smoke(argv):
    """The CI smoke test (reference deploy-kubernetes.yml:206-271): POST the
    sample request, require HTTP 200 + schema-valid body."""
    p = argparse.ArgumentParser(prog="creditcore smoke")
    p.add_argument("--url", default="http://127.0.0.1:5000")
    a = p.parse_args(argv)
    import httpx

    from .schema import SAMPLE_REQUEST, ModelOutput

    r = httpx.post(f"{a.url}/predict", json=SAMPLE_REQUEST, timeout=30.0)
    assert r.status_code == 200, f"smoke test failed: HTTP {r.status_code}: {r.text}"
    ModelOutput.model_validate(r.json())
    print(json.dumps(r.json()))
    print("SMOKE OK")


def main():
    cmds = {"train": _cmd_train, "pack": _cmd_pack, "serve": _cmd_serve, "smoke": _cmd_smoke}
    if len(sys.argv) < 2 or sys.argv[1] not in cmds:
        print(f"usage: python -m creditcore {{{'|'.join(cmds)}}} [args]", file=sys.stderr)
        sys.exit(2)
    cmds[sys.argv[1]](sys.argv[2:])


if __name__ == "__main__":
    main()
