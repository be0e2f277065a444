"""CLI entry: python -m kllms_amd <serve|bench|demo> [args...]

- serve: OpenAI-compatible HTTP server over the local engine
  (python -m kllms_amd serve --model llama-3-8b --port 8000)
- bench: the repo's driver-contract benchmark (bench.py passthrough)
- demo:  the examples/demo.py walkthrough on a tiny CPU model
"""

import os
import sys


def main() -> None:
    args = sys.argv[1:]
    cmd = args[0] if args else "help"
    rest = args[1:]
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    if cmd == "serve":
        from .server import main as serve_main

        serve_main(rest)
    elif cmd == "bench":
        sys.argv = [os.path.join(repo, "bench.py")] + rest
        with open(sys.argv[0]) as f:
            code = compile(f.read(), sys.argv[0], "exec")
        exec(code, {"__name__": "__main__", "__file__": sys.argv[0]})
    elif cmd == "demo":
        sys.argv = [os.path.join(repo, "examples", "demo.py")] + rest
        with open(sys.argv[0]) as f:
            code = compile(f.read(), sys.argv[0], "exec")
        exec(code, {"__name__": "__main__", "__file__": sys.argv[0]})
    else:
        print(__doc__)
        sys.exit(0 if cmd in ("help", "-h", "--help") else 2)


if __name__ == "__main__":
    main()
