"""Project-root templates: main.go, go.mod, Makefile, Dockerfile, README.

Parity targets: reference templates/main.go:22-282 (Main + MainUpdater),
templates/gomod.go:23-66, templates/makefile.go, templates/dockerfile.go,
templates/readme.go.

Note on the reference mainTemplate's unused
sigs.k8s.io/controller-runtime/pkg/controller import (main.go:178):
machinery's imports.Process (goimports) removes unused imports at
scaffold time, so the reference's real output never contains it; this
module omits it to the same effect, and the scaffold pipeline's own
format_go would strip it anyway (PARITY.md "resolved with evidence").
"""

from __future__ import annotations

from ..context import Context, hash_fnv
from ..machinery import File, Fragments, IfExists, Marker

IMPORT_MARKER = Marker("//", "kubebuilder:scaffold:imports")
SCHEME_MARKER = Marker("//", "kubebuilder:scaffold:scheme")
RECONCILER_MARKER = Marker("//", "kubebuilder:scaffold:reconcilers")

# pinned dependency versions for the generated project's go.mod
# (reference gomod.go:23-45; pinned per operator-builder issue #250)
GO_MOD_DEPENDENCIES = {
    "github.com/go-logr/logr": "v0.4.0",
    "github.com/nukleros/operator-builder-tools": "v0.2.0",
    "github.com/onsi/ginkgo": "v1.16.4",
    "github.com/onsi/gomega": "v1.15.0",
    "github.com/spf13/cobra": "v1.2.1",
    "github.com/stretchr/testify": "v1.7.0",
    "gopkg.in/yaml.v2": "v2.4.0",
    "k8s.io/api": "v0.22.2",
    "k8s.io/apimachinery": "v0.22.2",
    "k8s.io/client-go": "v0.22.2",
    "sigs.k8s.io/controller-runtime": "v0.10.2",
    "sigs.k8s.io/kubebuilder/v3": "v3.2.0",
    "sigs.k8s.io/yaml": "v1.2.0",
}

CRD_OPTIONS = "crd:preserveUnknownFields=false,crdVersions=v1,trivialVersions=true"


def main_go(ctx: Context) -> File:
    content = f"""{ctx.boilerplate}

package main

import (
\t"flag"
\t"os"

\t// Import all Kubernetes client auth plugins (e.g. Azure, GCP, OIDC, etc.)
\t// to ensure that exec-entrypoint and run can make use of them.
\t_ "k8s.io/client-go/plugin/pkg/client/auth"

\t"k8s.io/client-go/rest"
\t"k8s.io/apimachinery/pkg/runtime"
\tutilruntime "k8s.io/apimachinery/pkg/util/runtime"
\tclientgoscheme "k8s.io/client-go/kubernetes/scheme"
\tctrl "sigs.k8s.io/controller-runtime"
\t"sigs.k8s.io/controller-runtime/pkg/log/zap"
\t"sigs.k8s.io/controller-runtime/pkg/healthz"
\t{IMPORT_MARKER}
)

type ReconcilerInitializer interface {{
\tGetName() string
\tSetupWithManager(ctrl.Manager) error
}}

var (
\tscheme = runtime.NewScheme()
\tsetupLog = ctrl.Log.WithName("setup")
)

func init() {{
\tutilruntime.Must(clientgoscheme.AddToScheme(scheme))

\t{SCHEME_MARKER}
}}

func main() {{
\tvar metricsAddr string

\tvar enableLeaderElection bool

\tvar probeAddr string

\tflag.StringVar(&metricsAddr, "metrics-bind-address", ":8080", "The address the metric endpoint binds to.")
\tflag.StringVar(&probeAddr, "health-probe-bind-address", ":8081", "The address the probe endpoint binds to.")
\tflag.BoolVar(&enableLeaderElection, "leader-elect", false,
\t\t"Enable leader election for controller manager. " +
\t\t"Enabling this will ensure there is only one active controller manager.")

\topts := zap.Options{{
\t\tDevelopment: true,
\t}}
\topts.BindFlags(flag.CommandLine)
\tflag.Parse()

\tctrl.SetLogger(zap.New(zap.UseFlagOptions(&opts)))

\t// only print a given warning the first time we receive it
\trest.SetDefaultWarningHandler(
\t\trest.NewWarningWriter(os.Stderr, rest.WarningWriterOptions{{
\t\t\tDeduplicate: true,
\t\t}}),
\t)

\tmgr, err := ctrl.NewManager(ctrl.GetConfigOrDie(), ctrl.Options{{
\t\tScheme:                 scheme,
\t\tMetricsBindAddress:     metricsAddr,
\t\tPort:                   9443,
\t\tHealthProbeBindAddress: probeAddr,
\t\tLeaderElection:         enableLeaderElection,
\t\tLeaderElectionID:       "{hash_fnv(ctx.repo)}.{ctx.domain}",
\t}})
\tif err != nil {{
\t\tsetupLog.Error(err, "unable to start manager")
\t\tos.Exit(1)
\t}}

\treconcilers := []ReconcilerInitializer{{
\t\t{RECONCILER_MARKER}
\t}}

\tfor _, reconciler := range reconcilers {{
\t\tif err = reconciler.SetupWithManager(mgr); err != nil {{
\t\t\tsetupLog.Error(err, "unable to create controller", "controller", reconciler.GetName())
\t\t\tos.Exit(1)
\t\t}}
\t}}

\tif err := mgr.AddHealthzCheck("healthz", healthz.Ping); err != nil {{
\t\tsetupLog.Error(err, "unable to set up health check")
\t\tos.Exit(1)
\t}}

\tif err := mgr.AddReadyzCheck("readyz", healthz.Ping); err != nil {{
\t\tsetupLog.Error(err, "unable to set up ready check")
\t\tos.Exit(1)
\t}}

\tsetupLog.Info("starting manager")

\tif err := mgr.Start(ctrl.SetupSignalHandler()); err != nil {{
\t\tsetupLog.Error(err, "problem running manager")
\t\tos.Exit(1)
\t}}
}}
"""
    return File("main.go", content, IfExists.OVERWRITE)


def main_updater(
    ctx: Context,
    *,
    wire_resource: bool = True,
    wire_controller: bool = True,
) -> Fragments:
    """Wire a new API + controller into main.go at the scaffold markers
    (reference MainUpdater, templates/main.go:44-160; the
    WireResource/WireController knobs mirror machinery's MainUpdater
    fields set at scaffolds/api.go:154-159)."""
    res = ctx.resource

    imports: list[str] = []
    add_scheme: list[str] = []
    setup: list[str] = []

    if wire_resource:
        imports.append(f'{res.import_alias} "{res.path}"\n')
        add_scheme.append(
            f"utilruntime.Must({res.import_alias}.AddToScheme(scheme))\n"
        )

    if wire_controller:
        if ctx.multi_group and res.group:
            imports.append(
                f'{res.group}controllers '
                f'"{ctx.repo}/controllers/{res.group}"\n'
            )
            setup.append(
                f"{res.group}controllers.New{res.kind}Reconciler(mgr),\n"
            )
        else:
            imports.append(f'"{ctx.repo}/controllers"\n')
            setup.append(f"controllers.New{res.kind}Reconciler(mgr),\n")

    fragments: dict = {}
    if imports:
        fragments[IMPORT_MARKER] = imports
    if add_scheme:
        fragments[SCHEME_MARKER] = add_scheme
    if setup:
        fragments[RECONCILER_MARKER] = setup

    return Fragments(path="main.go", fragments=fragments)


def go_mod(ctx: Context) -> File:
    """Byte-faithful rendering of the reference's goModTemplate
    (templates/gomod.go:55-66): text/template's `{{ range }}` loop
    leaves a tab-only separator line between entries and `{{ end -}}`
    leaves the closing paren indented — go.mod is not gofmt'ed, so the
    real output keeps that shape (verified by the parity oracle)."""
    deps = "".join(
        f'\n\t"{name}" {version}\n\t'
        for name, version in sorted(GO_MOD_DEPENDENCIES.items())
    )
    content = f"\nmodule {ctx.repo}\n\ngo 1.15\n\nrequire (\n\t{deps})\n"
    return File("go.mod", content, IfExists.OVERWRITE)


def makefile(ctx: Context) -> File:
    # without a companion CLI the skipped `{{ if ne .RootCmdName "" -}}`
    # branch still leaves its surrounding newlines in the reference's
    # raw render (3 blank lines after `endef`; parity-oracle verified)
    cli_target = "\n"
    if ctx.cli_root_command_name:
        cli_target = f"""# Build the companion CLI
build-cli:
\tgo build -o bin/{ctx.cli_root_command_name} cmd/{ctx.cli_root_command_name}/main.go
"""

    content = f"""
# Image URL to use all building/pushing image targets
IMG ?= controller:latest
# Produce CRDs that work back to Kubernetes 1.11 (no version conversion)
CRD_OPTIONS ?= "{CRD_OPTIONS}"

# Get the currently used golang install path (in GOPATH/bin, unless GOBIN is set)
ifeq (,$(shell go env GOBIN))
GOBIN=$(shell go env GOPATH)/bin
else
GOBIN=$(shell go env GOBIN)
endif

# Setting SHELL to bash allows bash commands to be executed by recipes.
# This is a requirement for 'setup-envtest.sh' in the test target.
# Options are set to exit when a recipe line exits non-zero or a piped command fails.
SHELL = /usr/bin/env bash -o pipefail
.SHELLFLAGS = -ec

all: build

##@ General

# The help target prints out all targets with their descriptions organized
# beneath their categories. The categories are represented by '##@' and the
# target descriptions by '##'. The awk commands is responsible for reading the
# entire set of makefiles included in this invocation, looking for lines of the
# file as xyz: ## something, and then pretty-format the target and help. Then,
# if there's a line with ##@ something, that gets pretty-printed as a category.
# More info on the usage of ANSI control characters for terminal formatting:
# https://en.wikipedia.org/wiki/ANSI_escape_code#SGR_parameters
# More info on the awk command:
# http://linuxcommand.org/lc3_adv_awk.php

help: ## Display this help.
\t@awk 'BEGIN {{FS = ":.*##"; printf "\\nUsage:\\n  make \\033[36m<target>\\033[0m\\n"}} /^[a-zA-Z_0-9-]+:.*?##/ {{ printf "  \\033[36m%-15s\\033[0m %s\\n", $$1, $$2 }} /^##@/ {{ printf "\\n\\033[1m%s\\033[0m\\n", substr($$0, 5) }} ' $(MAKEFILE_LIST)

##@ Development

manifests: controller-gen ## Generate WebhookConfiguration, ClusterRole and CustomResourceDefinition objects.
\t$(CONTROLLER_GEN) $(CRD_OPTIONS) rbac:roleName=manager-role webhook paths="./..." output:crd:artifacts:config=config/crd/bases

generate: controller-gen ## Generate code containing DeepCopy, DeepCopyInto, and DeepCopyObject method implementations.
\t$(CONTROLLER_GEN) object:headerFile="hack/boilerplate.go.txt" paths="./..."

fmt: ## Run go fmt against code.
\tgo fmt ./...

vet: ## Run go vet against code.
\tgo vet ./...

ENVTEST_ASSETS_DIR=$(shell pwd)/testbin
test: manifests generate fmt vet ## Run tests.
\tmkdir -p ${{ENVTEST_ASSETS_DIR}}
\ttest -f ${{ENVTEST_ASSETS_DIR}}/setup-envtest.sh || curl -sSLo ${{ENVTEST_ASSETS_DIR}}/setup-envtest.sh https://raw.githubusercontent.com/kubernetes-sigs/controller-runtime/v0.7.2/hack/setup-envtest.sh
\tsource ${{ENVTEST_ASSETS_DIR}}/setup-envtest.sh; fetch_envtest_tools $(ENVTEST_ASSETS_DIR); setup_envtest_env $(ENVTEST_ASSETS_DIR); go test ./... -coverprofile cover.out

test-e2e:
\tgo test {ctx.repo}/test/e2e -tags=e2e_test -count=1

##@ Build

build: generate fmt vet ## Build manager binary.
\tgo build -o bin/manager main.go

run: manifests generate fmt vet ## Run a controller from your host.
\tgo run ./main.go

docker-build: test ## Build docker image with the manager.
\tdocker build -t ${{IMG}} .

docker-push: ## Push docker image with the manager.
\tdocker push ${{IMG}}

##@ Deployment

install: manifests kustomize ## Install CRDs into the K8s cluster specified in ~/.kube/config.
\t$(KUSTOMIZE) build config/crd | kubectl apply -f -

uninstall: manifests kustomize ## Uninstall CRDs from the K8s cluster specified in ~/.kube/config.
\t$(KUSTOMIZE) build config/crd | kubectl delete -f -

deploy: manifests kustomize ## Deploy controller to the K8s cluster specified in ~/.kube/config.
\tcd config/manager && $(KUSTOMIZE) edit set image controller=${{IMG}}
\t$(KUSTOMIZE) build config/default | kubectl apply -f -

undeploy: ## Undeploy controller from the K8s cluster specified in ~/.kube/config.
\t$(KUSTOMIZE) build config/default | kubectl delete -f -


CONTROLLER_GEN = $(shell pwd)/bin/controller-gen
controller-gen: ## Download controller-gen locally if necessary.
\t$(call go-get-tool,$(CONTROLLER_GEN),sigs.k8s.io/controller-tools/cmd/controller-gen@v0.4.1)

KUSTOMIZE = $(shell pwd)/bin/kustomize
kustomize: ## Download kustomize locally if necessary.
\t$(call go-get-tool,$(KUSTOMIZE),sigs.k8s.io/kustomize/kustomize/v3@v3.8.7)

# go-get-tool will 'go get' any package $2 and install it to $1.
PROJECT_DIR := $(shell dirname $(abspath $(lastword $(MAKEFILE_LIST))))
define go-get-tool
@[ -f $(1) ] || {{ \\
set -e ;\\
TMP_DIR=$$(mktemp -d) ;\\
cd $$TMP_DIR ;\\
go mod init tmp ;\\
echo "Downloading $(2)" ;\\
GOBIN=$(PROJECT_DIR)/bin go get $(2) ;\\
rm -rf $$TMP_DIR ;\\
}}
endef

{cli_target}
# Build the API Documentation
# NOTE: requires go version 1.16 or later
docs: manifests
\t@if ! command -v go &> /dev/null; then echo "error: go not installed"; exit 1; fi; \\
\tGOCMD=$$(which go); \\
\tif [[ -z $$($$GOCMD version | grep '1.16') ]]; then echo "error: requires go version >= 1.16"; exit 1; fi; \\
\tgo get fybrik.io/crdoc@v0.5.0; \\
\tgo install fybrik.io/crdoc@v0.5.0; \\
\tcrdoc --resources config/crd/bases/ --output docs/apis.md
"""
    return File("Makefile", content, IfExists.OVERWRITE)


def dockerfile(ctx: Context) -> File:
    content = """# Build the manager binary
FROM golang:1.16 as builder

WORKDIR /workspace
# Copy the Go Modules manifests
COPY go.mod go.mod
COPY go.sum go.sum
# cache deps before building and copying source so that we don't need to re-download as much
# and so that source changes don't invalidate our downloaded layer
RUN go mod download

# Copy the go source
COPY main.go main.go
COPY apis/ apis/
COPY controllers/ controllers/
COPY internal/ internal/

# Build
RUN CGO_ENABLED=0 GOOS=linux GOARCH=amd64 go build -a -o manager main.go

# Use distroless as minimal base image to package the manager binary
# Refer to https://github.com/GoogleContainerTools/distroless for more details
FROM gcr.io/distroless/static:nonroot
WORKDIR /
COPY --from=builder /workspace/manager .
USER 65532:65532

ENTRYPOINT ["/manager"]
"""
    return File("Dockerfile", content, IfExists.OVERWRITE)


def readme(ctx: Context) -> File:
    cli_section = ""
    if ctx.cli_root_command_name:
        cli_section = f"""## Companion CLI

To build the companion CLI:

    make build-cli

The CLI binary will get saved to the bin directory.  You can see the help
message with:

    ./bin/{ctx.cli_root_command_name} help
"""

    content = f"""A Kubernetes operator built with
[operator-builder](https://github.com/vmware-tanzu-labs/operator-builder).

## Local Development & Testing

To install the custom resource/s for this operator, make sure you have a
kubeconfig set up for a test cluster, then run:

    make install

To run the controller locally against a test cluster:

    make run

You can then test the operator by creating the sample manifest/s:

    kubectl apply -f config/samples

To clean up:

    make uninstall

## Deploy the Controller Manager

First, set the image:

    export IMG=myrepo/myproject:v0.1.0

Now you can build and push the image:

    make docker-build
    make docker-push

Then deploy:

    make deploy

To clean up:

    make undeploy

{cli_section}"""
    return File("README.md", content, IfExists.OVERWRITE)
