"""Hello world: BASELINE config 1 shape (CPU plumbing).

Run:  modal-amd run examples/hello.py::app.main
"""

import modal_amd as modal

app = modal.App("example-hello")


@app.function()
def square(x: int) -> int:
    return x * x


@app.local_entrypoint()
def main(n: int = 10):
    print("one call:", square.remote(7))
    print("fan-out:", sum(square.map(range(n))))
