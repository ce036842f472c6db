"""Job: legacy-style binding of op args per output stream (parity:
python/scannerpy/job.py). The modern path passes per-stream args via
`stream_args=` on ops and stream lists on Input/Output."""


class Job:
    def __init__(self, op_args):
        self.op_args = op_args
