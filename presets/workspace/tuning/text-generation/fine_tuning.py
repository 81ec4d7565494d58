"""Tuning preset entrypoint — reference layout parity with
presets/workspace/tuning/text-generation/fine_tuning.py (the image the
tuning Job runs, preset_tuning.go:145). Thin wrapper over the native
trainer in kaito_amd.tuning.fine_tuning (YAML-config LoRA/QLoRA SFT,
DDP over RCCL, peft-format adapter output + completion marker)."""
import sys

from kaito_amd.tuning.fine_tuning import main

if __name__ == "__main__":
    sys.exit(main())
