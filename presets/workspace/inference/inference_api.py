"""Preset image entrypoint (reference layout parity:
presets/workspace/inference/vllm/inference_api.py). Delegates to the
kaito_amd server entrypoint."""
from kaito_amd.server.entrypoint import main

if __name__ == "__main__":
    main()
