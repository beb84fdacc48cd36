#!/usr/bin/env python3
"""Validating admission webhook for DriverUpgradePolicySpec.

Consumer operators register this as a ValidatingWebhookConfiguration for
their CRs (e.g. AMDGPUDriver): any create/update whose
``spec.driverUpgradePolicy`` fails the API types' validation is rejected at
admission time with field-level messages, instead of surfacing later as a
reconcile error.

    python examples/policy_webhook.py --port 8443

POST /validate with an AdmissionReview v1; responds allowed=true/false.
"""

import argparse
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from pydantic import ValidationError

from k8s_operator_libs_amd.api.upgrade.v1alpha1 import DriverUpgradePolicySpec


def validate_policy(policy: dict) -> list:
    """Return a list of field-level error strings (empty = valid)."""
    try:
        DriverUpgradePolicySpec.model_validate(policy)
        return []
    except ValidationError as exc:
        return [
            f"{'.'.join(str(p) for p in err['loc'])}: {err['msg']}"
            for err in exc.errors()
        ]


def review_response(review: dict, policy_path=("spec", "driverUpgradePolicy")) -> dict:
    request = review.get("request", {}) or {}
    obj = request.get("object", {}) or {}
    node = obj
    for part in policy_path:
        node = (node or {}).get(part)
    errors = validate_policy(node) if node is not None else []
    response = {
        "uid": request.get("uid", ""),
        "allowed": not errors,
    }
    if errors:
        response["status"] = {
            "code": 422,
            "message": "invalid driverUpgradePolicy: " + "; ".join(errors),
        }
    return {
        "apiVersion": "admission.k8s.io/v1",
        "kind": "AdmissionReview",
        "response": response,
    }


def create_app():
    from fastapi import FastAPI, Request

    app = FastAPI(title="amd-upgrade-policy-webhook")

    @app.post("/validate")
    async def validate(request: Request):
        import json

        review = json.loads(await request.body())
        return review_response(review)

    return app


def main(argv=None) -> int:
    import uvicorn

    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("--host", default="0.0.0.0")
    parser.add_argument("--port", type=int, default=8443)
    parser.add_argument("--cert", help="TLS certificate (required by real apiservers)")
    parser.add_argument("--key", help="TLS private key")
    args = parser.parse_args(argv)
    uvicorn.run(create_app(), host=args.host, port=args.port,
                ssl_certfile=args.cert, ssl_keyfile=args.key)
    return 0


if __name__ == "__main__":
    sys.exit(main())
