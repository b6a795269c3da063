"""Doctest lane: the reference runs --doctest-modules across the whole
package (pytest.ini:1-15 there); here the doctest-bearing modules are
enumerated so the lane stays fast and GPU-free."""
import doctest

import pytest

import gordo_amd
import gordo_amd.core.import_utils
import gordo_amd.core.sensor_tag
import gordo_amd.machine.model.factories.feedforward_autoencoder
import gordo_amd.machine.model.factories.lstm_autoencoder
import gordo_amd.machine.model.factories.utils
import gordo_amd.machine.model.models
import gordo_amd.machine.model.register
import gordo_amd.machine.validators
import gordo_amd.reporters.mlflow
import gordo_amd.serializer.from_definition
import gordo_amd.serializer.into_definition
import gordo_amd.serializer.serializer
import gordo_amd.serializer.utils
import gordo_amd.server.properties
import gordo_amd.server.server
import gordo_amd.server.utils
import gordo_amd.util.text
import gordo_amd.util.utils
import gordo_amd.util.version
import gordo_amd.utils
import gordo_amd.workflow.workflow_generator.helpers

import sys

MODULES = [
    gordo_amd,
    gordo_amd.core.import_utils,
    gordo_amd.core.sensor_tag,
    gordo_amd.machine.model.factories.feedforward_autoencoder,
    gordo_amd.machine.model.factories.lstm_autoencoder,
    gordo_amd.machine.model.factories.utils,
    gordo_amd.machine.model.models,
    gordo_amd.machine.model.register,
    gordo_amd.machine.validators,
    gordo_amd.reporters.mlflow,
    sys.modules['gordo_amd.serializer.from_definition'],
    sys.modules['gordo_amd.serializer.into_definition'],
    gordo_amd.serializer.serializer,
    gordo_amd.serializer.utils,
    gordo_amd.server.properties,
    gordo_amd.server.server,
    gordo_amd.server.utils,
    gordo_amd.util.text,
    gordo_amd.util.utils,
    gordo_amd.util.version,
    gordo_amd.utils,
    gordo_amd.workflow.workflow_generator.helpers,
]


@pytest.mark.parametrize("module", MODULES, ids=lambda m: m.__name__)
def test_doctests(module):
    results = doctest.testmod(module, verbose=False)
    assert results.failed == 0, f"{results.failed} doctest failures in {module.__name__}"
