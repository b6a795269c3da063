"""
Anomaly blueprint — ``POST /gordo/v0/<project>/<name>/anomaly/prediction``
(behavioral spec: gordo/server/blueprints/anomaly.py).
"""
from __future__ import annotations

import io
import logging
import timeit
from typing import Any, Dict, Optional

from flask import Blueprint, g, jsonify, make_response, request, send_file

from .. import properties
from .. import utils

logger = logging.getLogger(__name__)

anomaly_blueprint = Blueprint("ioc_anomaly_blueprint", __name__)

DELETED_FROM_RESPONSE_COLUMNS = (
    "smooth-tag-anomaly-scaled",
    "smooth-total-anomaly-scaled",
    "smooth-tag-anomaly-unscaled",
    "smooth-total-anomaly-unscaled",
)


def _create_anomaly_response(start_time: Optional[float] = None):
    if start_time is None:
        start_time = timeit.default_timer()

    if g.y is None:
        return make_response(
            (
                jsonify(
                    {"message": "Cannot perform anomaly without 'y' to compare against."}
                ),
                400,
            )
        )

    try:
        anomaly_df = g.model.anomaly(g.X, g.y, frequency=properties.get_frequency())
    except AttributeError:
        msg = {
            "message": f"Model is not an AnomalyDetector, it is of type: {type(g.model)}"
        }
        return make_response(jsonify(msg), 422)

    # smoothed columns are dropped unless ?all_columns is given
    if request.args.get("all_columns") is None:
        columns_for_delete = [
            column
            for column in anomaly_df
            if column[0] in DELETED_FROM_RESPONSE_COLUMNS
        ]
        anomaly_df = anomaly_df.drop(columns=columns_for_delete)

    if request.args.get("format") == "parquet":
        return send_file(
            io.BytesIO(utils.dataframe_into_parquet_bytes(anomaly_df)),
            mimetype="application/octet-stream",
        )
    context: Dict[Any, Any] = {}
    context["time-seconds"] = f"{timeit.default_timer() - start_time:.4f}"
    return utils.frame_json_response(
        context, anomaly_df, context.pop("status-code", 200)
    )


@anomaly_blueprint.route(
    "/gordo/v0/<gordo_project>/<gordo_name>/anomaly/prediction", methods=["POST"]
)
@utils.model_required
@utils.extract_X_y
def post_anomaly_prediction():
    """POST X and y → anomaly response frame (tag/total anomaly columns,
    confidence vs thresholds)."""
    return _create_anomaly_response(timeit.default_timer())
