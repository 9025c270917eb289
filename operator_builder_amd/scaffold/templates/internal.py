"""User-editable stub templates: dependencies + mutate hooks, and the
config/ kustomize + sample templates.

Parity targets: reference templates/int/dependencies/component.go,
templates/int/mutate/component.go, templates/config/crd/kustomization.go,
templates/config/samples/crd_sample.go.
"""

from __future__ import annotations

import os

from ...utils import to_file_name
from ...workload.api_fields import APIFields
from ..context import Context
from ..machinery import File, Fragments, IfExists, Marker

CRD_RESOURCE_MARKER = Marker("#", "kubebuilder:scaffold:crdkustomizeresource")
CRD_WEBHOOK_MARKER = Marker("#", "kubebuilder:scaffold:crdkustomizewebhookpatch")
CRD_CAINJECTION_MARKER = Marker(
    "#", "kubebuilder:scaffold:crdkustomizecainjectionpatch"
)


def dependencies_component(ctx: Context) -> File:
    kind = ctx.resource.kind
    content = f"""{ctx.boilerplate}

package dependencies

import (
\t"github.com/nukleros/operator-builder-tools/pkg/controller/workload"
)

// {kind}CheckReady performs the logic to determine if a {kind} object is ready.
func {kind}CheckReady(r workload.Reconciler, req *workload.Request) (bool, error) {{
\treturn true, nil
}}
"""
    path = os.path.join(
        "internal", "dependencies", f"{to_file_name(kind)}.go"
    )
    return File(path, content, IfExists.SKIP)


def mutate_component(ctx: Context) -> File:
    kind = ctx.resource.kind
    content = f"""{ctx.boilerplate}

package mutate

import (
\t"github.com/nukleros/operator-builder-tools/pkg/controller/workload"
\t"sigs.k8s.io/controller-runtime/pkg/client"
)

// {kind}Mutate performs the logic to mutate resources that belong to the parent.
func {kind}Mutate(
\tr workload.Reconciler,
\treq *workload.Request,
\tobject client.Object,
) (replacedObjects []client.Object, skip bool, err error) {{
\treturn []client.Object{{object}}, false, nil
}}
"""
    path = os.path.join("internal", "mutate", f"{to_file_name(kind)}.go")
    return File(path, content, IfExists.SKIP)


CRD_KUSTOMIZATION_CONTENT = f"""# This kustomization.yaml is not intended to be run by itself,
# since it depends on service name and namespace that are out of this kustomize package.
# It should be run by config/default
resources:
{CRD_RESOURCE_MARKER}

patchesStrategicMerge:
# [WEBHOOK] To enable webhook, uncomment all the sections with [WEBHOOK] prefix.
# patches here are for enabling the conversion webhook for each CRD
{CRD_WEBHOOK_MARKER}

# [CERTMANAGER] To enable cert-manager, uncomment all the sections with [CERTMANAGER] prefix.
# patches here are for enabling the CA injection for each CRD
{CRD_CAINJECTION_MARKER}

# the following config is for teaching kustomize how to do kustomization for CRDs.
configurations:
- kustomizeconfig.yaml
"""


def crd_kustomization(ctx: Context) -> Fragments:
    """config/crd/kustomization.yaml creation + per-API base insertion
    (reference crd/kustomization.go)."""
    res = ctx.resource
    path = os.path.join("config", "crd", "kustomization.yaml")
    return Fragments(
        path=path,
        missing_file_content=CRD_KUSTOMIZATION_CONTENT,
        fragments={
            CRD_RESOURCE_MARKER: [
                f"- bases/{res.qualified_group}_{res.plural}.yaml\n"
            ],
            CRD_WEBHOOK_MARKER: [
                f"#- patches/webhook_in_{res.plural}.yaml\n"
            ],
            CRD_CAINJECTION_MARKER: [
                f"#- patches/cainjection_in_{res.plural}.yaml\n"
            ],
        },
    )


def crd_sample(
    ctx: Context, spec_fields: APIFields, is_cluster_scoped: bool
) -> File:
    res = ctx.resource
    lines = [
        f"apiVersion: {res.qualified_group}/{res.version}",
        f"kind: {res.kind}",
        "metadata:",
        f"  name: {res.kind.lower()}-sample",
    ]
    if not is_cluster_scoped:
        lines.append("  namespace: default")
    content = (
        "\n".join(lines)
        + "\n"
        + spec_fields.generate_sample_spec(required_only=False)
    )
    path = os.path.join(
        "config",
        "samples",
        f"{res.group}_{res.version}_{to_file_name(res.kind)}.yaml",
    )
    return File(path, content, IfExists.OVERWRITE)
