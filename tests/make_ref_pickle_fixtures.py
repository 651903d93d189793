"""Generate cross-serializer fixture payloads with the REFERENCE's vendored
cloudpickle (/root/reference/py/modal/_vendor/cloudpickle.py), standalone —
the reference package itself needs grpclib, but its pickler does not.

Run from the repo root: python tests/make_ref_pickle_fixtures.py
Writes tests/fixtures/ref_pickles/*.pkl; the companion test
(test_serialization.py::test_reference_cloudpickle_fixtures) deserializes
each with modal_amd's deserializer (SURVEY hard part 6: pickle byte-compat
with real cloudpickle clients)."""

from __future__ import annotations

import importlib.util
import os
import pickle
import sys
import types

OUT = os.path.join(os.path.dirname(__file__), "fixtures", "ref_pickles")


def load_ref_cloudpickle():
    # register under the REAL module path so fixtures carry the exact
    # global references a reference client's payload would
    modal_pkg = types.ModuleType("modal")
    modal_pkg.__path__ = []
    vendor_pkg = types.ModuleType("modal._vendor")
    vendor_pkg.__path__ = ["/root/reference/py/modal/_vendor"]
    sys.modules.setdefault("modal", modal_pkg)
    sys.modules.setdefault("modal._vendor", vendor_pkg)
    spec = importlib.util.spec_from_file_location(
        "modal._vendor.cloudpickle", "/root/reference/py/modal/_vendor/cloudpickle.py"
    )
    mod = importlib.util.module_from_spec(spec)
    sys.modules["modal._vendor.cloudpickle"] = mod
    spec.loader.exec_module(mod)
    return mod


def main() -> None:
    cp = load_ref_cloudpickle()
    os.makedirs(OUT, exist_ok=True)

    def write(name: str, data: bytes) -> None:
        with open(os.path.join(OUT, name), "wb") as f:
            f.write(data)
        print(f"{name}: {len(data)} bytes")

    # 1. a by-value function with a closure (the @app.function payload shape)
    factor = 7

    def closure_fn(x):
        return x * factor + 1

    write("closure_fn.pkl", cp.dumps(closure_fn))

    # 2. a lambda
    write("lambda.pkl", cp.dumps(lambda a, b=10: a + b))

    # 3. an exception with cause + traceback-ish payload (remote error shape)
    try:
        try:
            raise KeyError("inner-key")
        except KeyError as inner:
            raise ValueError("outer-message") from inner
    except ValueError as exc:
        write("exception.pkl", cp.dumps(exc))

    # 4. the reference FunctionInput.args form: pickled (args, kwargs)
    write("args_kwargs.pkl", cp.dumps(((1, "two", b"three"), {"k": [4, 5]})))

    # 5. nested containers with shared refs + recursion
    shared = {"deep": [1, 2, 3]}
    rec: list = [shared, shared]
    rec.append(rec)
    write("recursive.pkl", cp.dumps(rec))

    # 6. a by-value class with methods + instance state
    class Model:
        def __init__(self, w):
            self.w = w

        def predict(self, x):
            return self.w * x

    write("class_instance.pkl", cp.dumps(Model(3)))

    # 7. numpy payload (common user args)
    import numpy as np

    write("numpy.pkl", cp.dumps({"arr": np.arange(6).reshape(2, 3)}))

    # 8. GENERATOR_DONE-style sentinel data (plain dict form on the wire)
    write("plain_protocol4.pkl", pickle.dumps({"items_total": 5}, 4))


if __name__ == "__main__":
    main()
