#!/usr/bin/env python3
"""Serve a trained policy over HTTP (es_pytorch_amd/serve.py).

  python tools/serve_policy.py saved/<run>/weights/policy-40 --port 8080
  curl -X POST localhost:8080/act -H 'content-type: application/json' \
       -d '{"obs": [[0.0, 0.1, ...]]}'
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("policy", help="Policy pickle or torch.save'd module")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8080)
    args = p.parse_args()

    import uvicorn

    from es_pytorch_amd.serve import build_app
    uvicorn.run(build_app(args.policy), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
