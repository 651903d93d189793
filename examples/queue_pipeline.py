"""Queue producer/consumer across workers: BASELINE config 5 shape.

Run:  modal-amd run examples/queue_pipeline.py::app.main
"""

import modal_amd as modal

app = modal.App("example-queue")


@app.function()
def producer(q, n: int):
    for i in range(n):
        q.put(i * i)
    return n


@app.function()
def consumer(q, n: int):
    return sum(q.get(timeout=30) for _ in range(n))


@app.local_entrypoint()
def main(n: int = 100):
    with modal.Queue.ephemeral() as q:
        fc = consumer.spawn(q, n)
        producer.remote(q, n)
        print("sum of squares:", fc.get(timeout=60))
