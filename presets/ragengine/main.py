"""RAGEngine preset entrypoint (reference layout parity:
presets/ragengine/main.py)."""
from kaito_amd.ragengine.service import main

if __name__ == "__main__":
    main()
