"""Sandbox + Volume: BASELINE config 4 shape.

Run:  modal-amd run examples/sandbox_volume.py::app.main
"""

import io

import modal_amd as modal

app = modal.App("example-sandbox")


@app.local_entrypoint()
def main():
    vol = modal.Volume.from_name("example-vol", create_if_missing=True)
    with vol.batch_upload() as batch:
        batch.put_file(io.BytesIO(b"hello from the volume\n"), "greeting.txt")
    sb = modal.Sandbox.create("bash", "-c", "cat data/greeting.txt", volumes={"data": vol})
    sb.wait(raise_on_termination=False)
    print("sandbox said:", sb.stdout.read().strip())
