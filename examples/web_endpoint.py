"""Web endpoint served through the local HTTP gateway.

Run:  modal-amd serve examples/web_endpoint.py
"""

import modal_amd as modal

app = modal.App("example-web")


@app.function()
@modal.fastapi_endpoint(method="GET")
def greet(name: str = "world"):
    return {"hello": name}


@app.local_entrypoint()
def main():
    import urllib.request

    url = greet.web_url + "/?name=MI355X"
    print(url, "->", urllib.request.urlopen(url).read().decode())
