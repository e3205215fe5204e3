"""Write openapi.json from the gateway route table (reference:
clients/openapi-gen Makefile targets generate-openapi / generate-java-types).

Usage: python scripts/gen_openapi.py [out.json]
"""
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    from smg_amd.config import PolicyConfig, RouterConfig
    from smg_amd.server.app import build_app
    from smg_amd.server.app_context import AppContext
    from smg_amd.server.openapi import build_openapi

    cfg = RouterConfig(policy=PolicyConfig(name="round_robin", gpu_tree=False))
    app = build_app(AppContext(cfg))
    doc = build_openapi(app)
    out = sys.argv[1] if len(sys.argv) > 1 else "openapi.json"
    with open(out, "w") as f:
        json.dump(doc, f, indent=2)
    print(f"{out}: {len(doc['paths'])} paths, {len(doc['components']['schemas'])} schemas")


if __name__ == "__main__":
    main()
