"""Volume and Image behavior on the local block/CAS backends."""

from __future__ import annotations

import io
import os

import pytest

import modal_amd as modal
from modal_amd.exception import NotFoundError


def test_volume_upload_read_list(client, tmp_path):
    local = tmp_path / "src.bin"
    payload = os.urandom(300_000)
    local.write_bytes(payload)
    with modal.Volume.ephemeral() as vol:
        with vol.batch_upload() as batch:
            batch.put_file(str(local), "dir/src.bin")
            batch.put_file(io.BytesIO(b"inline"), "inline.txt")
        data = b"".join(chunk for chunk in vol.read_file("dir/src.bin"))
        assert data == payload
        entries = {e.path: e for e in vol.listdir("/", recursive=True)}
        assert entries["dir/src.bin"].size == len(payload)
        assert entries["inline.txt"].size == 6
        vol.remove_file("inline.txt")
        with pytest.raises(NotFoundError):
            b"".join(chunk for chunk in vol.read_file("inline.txt"))


def test_volume_large_file_blocks(client, tmp_path):
    """Files above the 8 MiB block size split into multiple CAS blocks."""
    payload = os.urandom(1024) * (9 * 1024)  # 9 MiB
    with modal.Volume.ephemeral() as vol:
        with vol.batch_upload() as batch:
            batch.put_file(io.BytesIO(payload), "big.bin")
        data = b"".join(chunk for chunk in vol.read_file("big.bin"))
        assert data == payload


def test_volume_copy_and_commit(client):
    with modal.Volume.ephemeral() as vol:
        with vol.batch_upload() as batch:
            batch.put_file(io.BytesIO(b"abc"), "a.txt")
        vol.copy_files(["a.txt"], "b.txt")
        assert b"".join(vol.read_file("b.txt")) == b"abc"
        vol.commit()
        vol.reload()


def test_volume_named_persistence(client):
    v1 = modal.Volume.from_name("train-data", create_if_missing=True)
    with v1.batch_upload() as batch:
        batch.put_file(io.BytesIO(b"persisted"), "x.txt")
    v2 = modal.Volume.from_name("train-data")
    assert b"".join(v2.read_file("x.txt")) == b"persisted"
    modal.Volume.delete("train-data")
    with pytest.raises(NotFoundError):
        modal.Volume.from_name("train-data").hydrate()


def test_volume_in_worker_function(client):
    """Functions see mounted volumes via the shared tree (config 4 path)."""
    app = modal.App("vol-app")
    vol = modal.Volume.from_name("fn-vol", create_if_missing=True)

    @app.function(volumes={"/data-vol": vol})
    def writer(text):
        with open("/data-vol/out.txt", "w") as f:
            f.write(text)
        return "written"

    with app.run(client=client):
        assert writer.remote("from-worker") == "written"
    v2 = modal.Volume.from_name("fn-vol")
    assert b"".join(v2.read_file("out.txt")) == b"from-worker"


def test_image_recipe_dedup_and_build(client):
    img1 = modal.Image.debian_slim().env({"FOO": "1"}).run_commands("echo built > marker.txt")
    img2 = modal.Image.debian_slim().env({"FOO": "1"}).run_commands("echo built > marker.txt")
    img1.hydrate()
    img2.hydrate()
    assert img1.object_id == img2.object_id  # content-addressed recipe dedup
    img3 = modal.Image.debian_slim().env({"FOO": "2"})
    img3.hydrate()
    assert img3.object_id != img1.object_id
    assert "echo built" in img1.build_log()


def test_image_workdir_env_dockerfile(client, tmp_path):
    df = tmp_path / "Dockerfile"
    df.write_text("FROM scratch\nENV A=b\nWORKDIR /work\nRUN echo hi\n")
    img = modal.Image.from_dockerfile(str(df))
    img.hydrate()
    assert img.object_id.startswith("im-")


def test_image_pip_install_importable_ok(client):
    # numpy is importable in the base interpreter: the layer builds offline
    img = modal.Image.debian_slim().pip_install("numpy")
    img.hydrate()
    assert img.object_id.startswith("im-")


def test_image_on_function(client):
    app = modal.App("img-app")
    img = modal.Image.debian_slim().env({"LAYER_VAR": "present"})

    @app.function(image=img)
    def read_env():
        import os

        return os.environ.get("LAYER_VAR")

    with app.run(client=client):
        assert read_env.remote() == "present"


def test_mount_python_packages(client):
    m = modal.Mount.from_local_python_packages("modal_amd.utils")
    m.hydrate()
    assert m.object_id.startswith("mo-")


def test_image_run_function_executes(client, tmp_path):
    marker = str(tmp_path / "built-by-fn")

    def build_step(path):
        with open(path, "w") as f:
            f.write("ran at build time")

    img = modal.Image.debian_slim().run_function(build_step, marker)
    img.hydrate()
    assert open(marker).read() == "ran at build time"


def test_image_run_function_failure_surfaces(client):
    def bad_step():
        raise RuntimeError("build exploded")

    img = modal.Image.debian_slim().run_function(bad_step)
    from modal_amd.exception import ExecutionError

    with pytest.raises(ExecutionError, match="build exploded|run_function"):
        img.hydrate()


def test_exit_hook_writes_land_before_volume_commit(client):
    """@exit hooks run before the exit-time volume commit (parity:
    lifecycle finalization then task_lifecycle_manager.py:117-120), so
    files written in @exit are visible in the committed volume."""
    app = modal.App("exit-vol-app")
    vol = modal.Volume.from_name("exit-vol", create_if_missing=True)

    @app.cls(volumes={"/exit-vol": vol})
    class Svc:
        @modal.enter()
        def up(self):
            self.n = 0

        @modal.method()
        def bump(self):
            self.n += 1
            return self.n

        @modal.exit()
        def down(self):
            with open("/exit-vol/final.txt", "w") as f:
                f.write(f"count={self.n}")

    with app.run(client=client):
        assert Svc().bump.remote() == 1
    v2 = modal.Volume.from_name("exit-vol")
    assert b"".join(v2.read_file("final.txt")) == b"count=1"


def test_image_shell_override(client, tmp_path, run_dir):
    """Image.shell() sets the SHELL for later run_commands (parity:
    reference _image.py:1990). With isolation on, absolute-path writes
    land in the image's fs layer, so look there first."""
    import os

    marker = tmp_path / "shellname.txt"
    img = (
        modal.Image.debian_slim()
        .shell(["/bin/bash", "-c"])
        .run_commands(f"echo $0 > {marker}")
    )
    img.hydrate()
    candidates = [str(marker)]
    images_root = os.path.join(run_dir, "images")
    if os.path.isdir(images_root):
        candidates += [
            os.path.join(images_root, d, "fsdiff", str(marker).lstrip("/"))
            for d in os.listdir(images_root)
        ]
    hits = [p for p in candidates if os.path.exists(p)]
    assert hits, "run_commands produced no marker file"
    assert "bash" in open(hits[0]).read()
