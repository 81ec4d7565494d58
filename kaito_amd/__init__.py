"""kaito_amd — MI355X-native LLM operator + inference engine.

Brand-new framework with the capabilities of kaito-project/kaito
(reference at /root/reference), built MI355X-first: hand-written CDNA4 HIP
kernels (MFMA/LDS) for the serving hot path, RCCL over xGMI for TP, and a
Python operator/planning layer (SKU table, estimator, parallelism planner,
manifest generation) mirroring the reference's Go controllers.
"""
__version__ = "0.1.0"
