"""Control-plane message types between driver and trial workers.

Same conceptual protocol as the reference RPC layer
(/root/reference/maggy/core/rpc.py:307-333): REG / TRIAL(=GET answer) /
METRIC (via shared-memory ring, not a message) / FINAL / STOP (via shared
stop word) / GSTOP / BLACK (driver-internal requeue on worker death).
"""
REG = "REG"        # worker -> driver: ready for work
TRIAL = "TRIAL"    # driver -> worker: (trial_id, params)
FINAL = "FINAL"    # worker -> driver: (trial_id, opt_val, duration, early, logs)
ERROR = "ERROR"    # worker -> driver: (trial_id, traceback)
GSTOP = "GSTOP"    # driver -> worker: experiment done, exit
LOG = "LOG"        # worker -> driver: mid-trial log text (throttled)
