"""Time-series forecast protocol (optional /timeseries routes).

Reference parity: python/kserve protocol/rest/timeseries/{endpoints,
dataplane,types}.py (~450 LoC): POST /timeseries/v1/forecast against models
exposing a ``forecast`` method.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from fastapi import APIRouter, Request
from fastapi.responses import JSONResponse
from pydantic import BaseModel, Field

from kserve_amd.errors import ModelNotFound, ModelNotReady
from kserve_amd.protocol.dataplane import DataPlane


class ForecastRequest(BaseModel):
    model: str
    inputs: List[Dict[str, Any]]  # series: {"timestamps": [...], "values": [...]}
    horizon: int = 1
    quantiles: Optional[List[float]] = None
    parameters: Dict[str, Any] = Field(default_factory=dict)


class ForecastResponse(BaseModel):
    model: str
    outputs: List[Dict[str, Any]]


class TimeSeriesModelMixin:
    """Models implementing ``forecast(series, horizon, quantiles, params)``."""

    async def forecast(
        self,
        inputs: List[Dict[str, Any]],
        horizon: int,
        quantiles: Optional[List[float]] = None,
        parameters: Optional[Dict[str, Any]] = None,
    ) -> List[Dict[str, Any]]:
        raise NotImplementedError


class TimeSeriesEndpoints:
    def __init__(self, dataplane: DataPlane):
        self.dataplane = dataplane

    async def forecast(self, raw_request: Request):
        body = await raw_request.json()
        try:
            req = ForecastRequest.model_validate(body)
        except Exception as e:
            return JSONResponse(status_code=400, content={"error": str(e)})
        model = self.dataplane.get_model(req.model)
        if not hasattr(model, "forecast"):
            return JSONResponse(
                status_code=400,
                content={"error": f"model {req.model} does not support forecast"},
            )
        outputs = await model.forecast(
            req.inputs, req.horizon, req.quantiles, req.parameters
        )
        return JSONResponse(
            content=ForecastResponse(model=req.model, outputs=outputs).model_dump()
        )


def register_timeseries_endpoints(app, dataplane: DataPlane):
    ep = TimeSeriesEndpoints(dataplane)
    router = APIRouter(tags=["TimeSeries"])
    router.add_api_route("/timeseries/v1/forecast", ep.forecast, methods=["POST"])
    app.include_router(router)
    return ep
