

def tune_gc() -> None:
    """Serving-process GC posture: freeze startup objects out of
    collection and raise gen-0 thresholds so steady-state request churn
    (short-lived dicts/bytes per request) doesn't trigger full scans.
    Measured ~2-3% worker throughput on the 4k-token chat bench."""
    import gc

    gc.collect()
    gc.freeze()
    gc.set_threshold(50000, 50, 50)
