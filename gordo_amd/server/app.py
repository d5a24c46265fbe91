"""Model server.

Parity: gordo_components/server/server.py + views — one server per model dir:

    POST /prediction          X (JSON or parquet) → model-input/model-output
    POST /anomaly/prediction  adds tag-anomaly-scores / total-anomaly-score
    GET  /metadata            build metadata
    GET  /download-model      serialized model bytes
    GET  /healthcheck         liveness

MI355X-native: FastAPI/uvicorn (async) instead of Flask/gunicorn; the model is
loaded once; inference goes through ModelRuntime's fused HIP path on GPU.
Request/response formats match the reference's capability: JSON records,
nested dict, parquet, arrow IPC, or raw float32 (?format=raw + X-Gordo-Shape header).
"""
import logging

import numpy as np
import pandas as pd
from fastapi import FastAPI, Request, Response
from fastapi.responses import JSONResponse

from .. import __version__
from ..serializer import dumps
from .model_io import ModelRuntime
from .utils import (
    dataframe_from_arrow_bytes,
    dataframe_from_dict,
    dataframe_from_parquet_bytes,
    dataframe_into_arrow_bytes,
    dataframe_into_parquet_bytes,
    dataframe_to_dict,
)
from ..model.utils import make_base_dataframe

logger = logging.getLogger(__name__)


def build_app(model_dir: str, device=None):
    runtime = ModelRuntime(model_dir, device=device)
    app = FastAPI(title="gordo-amd model server", version=__version__)
    app.state.runtime = runtime

    # ------------------------------------------------------------------
    async def _extract_frames(request: Request):
        ctype = request.headers.get("content-type", "")
        body = await request.body()
        try:
            if "x-gordo-raw" in ctype:
                # zero-copy fast path: body is row-major float32 LE, shape
                # in the X-Gordo-Shape header ("N,F"); columns are
                # positional by construction (training tag order)
                shape = request.headers.get("x-gordo-shape", "")
                n, f = (int(v) for v in shape.split(","))
                arr = np.frombuffer(body, dtype="<f4").reshape(n, f)
                X = pd.DataFrame(arr)
                y = None
            elif "arrow" in ctype:
                X = dataframe_from_arrow_bytes(body)
                y = None
            elif "octet-stream" in ctype or "parquet" in ctype:
                X = dataframe_from_parquet_bytes(body)
                y = None
            else:
                import json

                payload = json.loads(body) if body else {}
                if not isinstance(payload, dict) or (
                    "X" not in payload and "data" not in payload
                ):
                    return None, None, JSONResponse(
                        {"error": "Body must contain 'X'"}, status_code=400
                    )
                Xp = payload.get("X", payload)
                X = dataframe_from_dict(Xp if isinstance(Xp, dict) else {"data": Xp})
                y = None
                if payload.get("y") is not None:
                    yp = payload["y"]
                    y = dataframe_from_dict(yp if isinstance(yp, dict) else {"data": yp})
        except Exception as e:
            # undecodable body (corrupt parquet/arrow, invalid JSON, ragged
            # arrays): a client error, not a server fault — the reference
            # 400s on bad payloads
            return None, None, JSONResponse(
                {"error": f"Could not decode request body as {ctype or 'json'}: {e}"},
                status_code=400,
            )
        # column sanity: when metadata lists tags and X has plain columns,
        # reorder to the training tag order. A width match alone is NOT
        # enough for a frame with real column names (a misordered parquet
        # frame would silently score garbage — the reference 400s on tag
        # mismatch): named-but-unknown columns require the explicit
        # X-Gordo-Assume-Tag-Order opt-in. Unnamed (RangeIndex) payloads —
        # plain JSON value arrays — are positional by construction and are
        # assigned the training tag order as the reference does.
        tags = runtime.tag_names
        if tags and not isinstance(X.columns, pd.MultiIndex):
            missing = [t for t in tags if t not in X.columns]
            positional = isinstance(X.columns, pd.RangeIndex) or all(
                isinstance(c, (int, np.integer)) for c in X.columns
            )
            assume_order = request.headers.get(
                "x-gordo-assume-tag-order", ""
            ).lower() in ("1", "true", "yes")
            if not missing:
                if list(X.columns) != tags:
                    X = X[tags]
            elif X.shape[1] != len(tags):
                return None, None, JSONResponse(
                    {
                        "error": f"Expected {len(tags)} tag columns, got "
                        f"{X.shape[1]}; missing: {missing[:5]}"
                    },
                    status_code=400,
                )
            elif positional or assume_order:
                X = X.copy(deep=False)
                X.columns = tags
            else:
                return None, None, JSONResponse(
                    {
                        "error": "Column names do not match the training tag "
                        f"list (missing: {missing[:5]}). Send columns named "
                        "by tag, or positional arrays, or set the "
                        "X-Gordo-Assume-Tag-Order: true header to assert the "
                        "columns are already in training order.",
                        "expected-tags": tags,
                    },
                    status_code=400,
                )
        return X, y, None

    def _respond(df: pd.DataFrame, request: Request) -> Response:
        fmt = request.query_params.get("format", "json")
        accept = request.headers.get("accept", "")
        if fmt == "raw" or "x-gordo-raw" in accept:
            # row-major float32 LE values; shape in X-Gordo-Shape, column
            # layout on request (?columns=1 — it is fixed per model, so
            # clients normally fetch it once). Values-only: the caller keeps
            # its own time index; lookback models emit fewer rows than sent
            # (rows align to the TAIL of the input, like every other format)
            headers = {"X-Gordo-Shape": f"{df.shape[0]},{df.shape[1]}"}
            if request.query_params.get("columns"):
                import json as _json

                headers["X-Gordo-Columns"] = _json.dumps(
                    [list(c) if isinstance(c, tuple) else [c, ""] for c in df.columns]
                )
            return Response(
                content=np.ascontiguousarray(df.to_numpy(np.float32)).tobytes(),
                media_type="application/x-gordo-raw",
                headers=headers,
            )
        if fmt == "arrow" or "arrow" in accept:
            return Response(
                content=dataframe_into_arrow_bytes(df),
                media_type="application/vnd.apache.arrow.stream",
            )
        if fmt == "parquet" or "octet-stream" in accept:
            return Response(
                content=dataframe_into_parquet_bytes(df),
                media_type="application/octet-stream",
            )
        return JSONResponse(dataframe_to_dict(df))

    # ------------------------------------------------------------------
    @app.get("/healthcheck")
    async def healthcheck():
        return {"gordo-server-version": __version__}

    @app.get("/metadata")
    async def metadata():
        return {"version": __version__, "metadata": runtime.metadata}

    @app.get("/download-model")
    async def download_model():
        return Response(content=dumps(runtime.model), media_type="application/octet-stream")

    @app.post("/prediction")
    async def prediction(request: Request):
        X, _, err = await _extract_frames(request)
        if err is not None:
            return err
        try:
            out = runtime.predict(X)
        except Exception as e:
            logger.exception("prediction failed")
            return JSONResponse({"error": str(e)}, status_code=422)
        df = make_base_dataframe(
            tags=list(X.columns),
            model_input=X.values,
            model_output=out,
            index=np.asarray(X.index),
        )
        return _respond(df, request)

    @app.post("/anomaly/prediction")
    async def anomaly_prediction(request: Request):
        if not runtime.is_anomaly:
            return JSONResponse(
                {"error": "Model is not an anomaly detector; use /prediction"},
                status_code=422,
            )
        X, y, err = await _extract_frames(request)
        if err is not None:
            return err
        freq = None
        ds_meta = runtime.metadata.get("dataset", {})
        if ds_meta.get("resolution"):
            try:
                freq = pd.Timedelta(ds_meta["resolution"])
            except ValueError:
                freq = None
        try:
            df = runtime.anomaly(X, y, frequency=freq)
        except Exception as e:
            logger.exception("anomaly failed")
            return JSONResponse({"error": str(e)}, status_code=422)
        return _respond(df, request)

    return app


def run_server(model_dir: str, host: str = "0.0.0.0", port: int = 5555, workers: int = 1):
    import uvicorn

    app = build_app(model_dir)
    uvicorn.run(app, host=host, port=port, workers=workers, log_level="info")
