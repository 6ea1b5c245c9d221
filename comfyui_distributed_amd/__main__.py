"""``python -m comfyui_distributed_amd`` — start the master (or worker)
server (reference entry: ComfyUI imports distributed.py at startup; this
framework is standalone)."""

from .server.app import main

if __name__ == "__main__":
    main()
