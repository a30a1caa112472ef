"""Fetch the current global model without participating: a plain GET /model
against the coordinator REST API (bincode body, 204 while no model exists).
"""
import sys

from xaynet_amd import _core


def main(url: str = "http://127.0.0.1:8081"):
    host, _, port = url.split("//")[-1].rpartition(":")
    client = _core.rest.HttpClient(host, int(port))
    result = client.request("GET", "/model")
    if result is None:
        print("coordinator unreachable")
        return 1
    status, body = result
    if status == 204:
        print("no global model yet (204)")
        return 0
    # decode as f32 (dtype 0); pass 1/2/3 for f64/i32/i64 per the round config
    model = _core.sdk.decode_model(b"\x01" + body, 0)
    print(f"global model ({len(model)} weights): {model[:8]}{'...' if len(model) > 8 else ''}")
    return 0


if __name__ == "__main__":
    sys.exit(main(*sys.argv[1:]))
