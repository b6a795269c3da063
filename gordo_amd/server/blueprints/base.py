"""
Base model-serving blueprint — routes under
``/gordo/v0/<project>/<name>/`` (behavioral spec:
gordo/server/blueprints/base.py).
"""
from __future__ import annotations

import io
import logging
import os
import timeit
import traceback
from typing import Any, Dict

import pandas as pd
from flask import (
    Blueprint,
    current_app,
    g,
    jsonify,
    make_response,
    request,
    send_file,
)

import gordo_amd
from ... import serializer
from ...machine.model import utils as model_utils
from .. import model_io
from .. import utils as server_utils
from ..properties import get_tags, get_target_tags

logger = logging.getLogger(__name__)

base_blueprint = Blueprint("base_model_view", __name__)


@base_blueprint.route(
    "/gordo/v0/<gordo_project>/<gordo_name>/prediction", methods=["POST"]
)
@server_utils.model_required
@server_utils.extract_X_y
def post_prediction():
    """POST X (JSON dict-of-dicts or parquet) → model-input/model-output
    response frame."""
    context: Dict[Any, Any] = {}
    X = g.X
    start_time_s = timeit.default_timer()
    try:
        output = model_io.get_model_output(model=g.model, X=X)
    except ValueError as err:
        logger.error(
            "Failed to predict or transform; error: %s - \nTraceback: %s",
            err, traceback.format_exc(),
        )
        context["error"] = f"ValueError: {str(err)}"
        return make_response((jsonify(context), 400))
    except Exception as exc:
        logger.error(
            "Failed to predict or transform; error: %s - \nTraceback: %s",
            exc, traceback.format_exc(),
        )
        context["error"] = "Something unexpected happened; check your input data"
        return make_response((jsonify(context), 400))

    logger.debug(
        "Calculating model output took %s s",
        timeit.default_timer() - start_time_s,
    )
    data = model_utils.make_base_dataframe(
        tags=get_tags(),
        model_input=X.values if isinstance(X, pd.DataFrame) else X,
        model_output=output,
        target_tag_list=get_target_tags(),
        index=X.index,
    )
    if request.args.get("format") == "parquet":
        return send_file(
            io.BytesIO(server_utils.dataframe_into_parquet_bytes(data)),
            mimetype="application/octet-stream",
        )
    return server_utils.frame_json_response(
        context, data, context.pop("status-code", 200)
    )


@base_blueprint.route(
    "/gordo/v0/<gordo_project>/<gordo_name>/revision/<revision>", methods=["DELETE"]
)
def delete_model_revision(gordo_name: str, revision: str, **kwargs):
    server_utils.validate_gordo_name(gordo_name)
    if not server_utils.validate_revision(revision):
        return make_response(
            (jsonify({"error": "Revision should only contains numbers."}), 422)
        )
    if revision == g.current_revision:
        return make_response(
            (jsonify({"error": "Unable to delete current revision."}), 409)
        )
    revision_dir = os.path.join(g.collection_dir, "..", revision)
    server_utils.delete_revision(revision_dir, gordo_name)
    return make_response(jsonify({"ok": True}), 200)


@base_blueprint.route(
    "/gordo/v0/<gordo_project>/<gordo_name>/metadata", methods=["GET"]
)
@base_blueprint.route(
    "/gordo/v0/<gordo_project>/<gordo_name>/healthcheck", methods=["GET"]
)
@server_utils.metadata_required
def get_metadata():
    """Model metadata; doubles as the per-model healthcheck."""
    model_collection_env_var = current_app.config["MODEL_COLLECTION_DIR_ENV_VAR"]
    metadata = dict(g.info) if g.info else {}
    metadata.update(
        {
            "gordo-server-version": gordo_amd.__version__,
            "metadata": g.metadata,
            "env": {
                model_collection_env_var: os.environ.get(model_collection_env_var)
            },
        }
    )
    return metadata


@base_blueprint.route(
    "/gordo/v0/<gordo_project>/<gordo_name>/download-model", methods=["GET"]
)
@server_utils.model_required
def get_download_model():
    serialized_model = serializer.dumps(g.model)
    return send_file(io.BytesIO(serialized_model), download_name="model.pickle")


@base_blueprint.route("/gordo/v0/<gordo_project>/models", methods=["GET"])
def get_model_list(gordo_project: str):
    try:
        # models are directories; stray files in the collection dir
        # (e.g. a fleet status JSON) are not models
        available_models = [
            name
            for name in os.listdir(g.collection_dir)
            if os.path.isdir(os.path.join(g.collection_dir, name))
        ]
    except FileNotFoundError:
        available_models = []
    return jsonify({"models": available_models})


@base_blueprint.route("/gordo/v0/<gordo_project>/revisions", methods=["GET"])
def get_revision_list(gordo_project: str):
    try:
        available_revisions = os.listdir(os.path.join(g.collection_dir, ".."))
    except FileNotFoundError:
        logger.error(
            "Attempted to list directories above %s but failed with: %s",
            g.collection_dir, traceback.format_exc(),
        )
        available_revisions = [g.current_revision]
    return jsonify(
        {
            # latest = what the server was deployed with; revision = the
            # one this request asked for (reference
            # test_gordo_server.py::test_list_revisions keys)
            "latest": g.current_revision,
            "revision": g.revision,
            "available-revisions": available_revisions,
        }
    )


@base_blueprint.route("/gordo/v0/<gordo_project>/expected-models", methods=["GET"])
def get_expected_models(gordo_project: str):
    return jsonify({"expected-models": current_app.config["EXPECTED_MODELS"]})
