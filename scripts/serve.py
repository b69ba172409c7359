"""Serve the pose-estimation pipeline over HTTP (see improved_body_parts_amd/serve.py).

    python scripts/serve.py --checkpoint checkpoints/PoseNet_52_epoch.pth
    curl -X POST --data-binary @image.png http://127.0.0.1:8000/pose
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--checkpoint", default=None)
    ap.add_argument("--config", default="Canonical")
    ap.add_argument("--nstack", type=int, default=4)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    args = ap.parse_args()

    from improved_body_parts_amd.serve import create_app
    import uvicorn
    app = create_app(config_name=args.config, nstack=args.nstack,
                     checkpoint=args.checkpoint)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
