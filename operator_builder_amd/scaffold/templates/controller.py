"""Controller templates: reconciler, phase registration, suite test.

Parity targets: reference templates/controller/controller.go:34-441,
templates/controller/phases.go:27-115,
templates/controller/controller_suitetest.go:31-171.
"""

from __future__ import annotations

import os

from ...utils import to_file_name
from ...workload.kinds import Workload
from ..context import Context
from ..machinery import File, Fragments, IfExists, Marker

SUITE_IMPORT_MARKER = Marker("//", "kubebuilder:scaffold:imports")
SUITE_SCHEME_MARKER = Marker("//", "kubebuilder:scaffold:scheme")


def _imports_block(ctx: Context, builder: Workload) -> str:
    base = ['"context"', '"fmt"']
    if builder.is_component():
        base += ['"errors"', '"reflect"']

    other = [
        '"github.com/go-logr/logr"',
        'apierrs "k8s.io/apimachinery/pkg/api/errors"',
        '"k8s.io/client-go/tools/record"',
        'ctrl "sigs.k8s.io/controller-runtime"',
        '"sigs.k8s.io/controller-runtime/pkg/client"',
        '"sigs.k8s.io/controller-runtime/pkg/controller"',
        '"github.com/nukleros/operator-builder-tools/pkg/controller/phases"',
        '"github.com/nukleros/operator-builder-tools/pkg/controller/predicates"',
        '"github.com/nukleros/operator-builder-tools/pkg/controller/workload"',
    ]
    if builder.is_component():
        other += [
            '"github.com/nukleros/operator-builder-tools/pkg/resources"',
            '"sigs.k8s.io/controller-runtime/pkg/event"',
            '"sigs.k8s.io/controller-runtime/pkg/handler"',
            '"sigs.k8s.io/controller-runtime/pkg/predicate"',
            '"sigs.k8s.io/controller-runtime/pkg/reconcile"',
            '"sigs.k8s.io/controller-runtime/pkg/source"',
            '"k8s.io/apimachinery/pkg/types"',
        ]

    res = ctx.resource
    internal = [
        f'"{ctx.repo}/internal/dependencies"',
        f'"{ctx.repo}/internal/mutate"',
        f'{res.import_alias} "{res.path}"',
    ]
    if builder.is_component():
        col = builder.get_collection()
        internal.append(
            f"{col.get_api_group()}{col.get_api_version()} "
            f'"{ctx.repo}/apis/{col.get_api_group()}/{col.get_api_version()}"'
        )
    if builder.has_child_resources():
        internal.append(f'"{res.path}/{builder.get_package_name()}"')

    lines = []
    for group_lines in (base, other, internal):
        for imp in group_lines:
            lines.append(f"\t{imp}\n")
        lines.append("\n")
    return "".join(lines).rstrip("\n") + "\n"


def controller(ctx: Context, builder: Workload) -> File:
    res = ctx.resource
    kind = res.kind

    rbac_markers = "".join(
        f"{rule.to_marker()}\n" for rule in builder.get_rbac_rules()
    )

    collection_requeue = ""
    if builder.is_component():
        collection_requeue = """\t\tif errors.Is(err, workload.ErrCollectionNotFound) {
\t\t\treturn ctrl.Result{Requeue: true}, nil
\t\t}

"""

    if builder.is_component():
        new_request_return = (
            "\t// store the collection and return any resulting error\n"
            "\treturn workloadRequest, r.SetCollection(component, workloadRequest)"
        )
    else:
        new_request_return = "\treturn workloadRequest, nil"

    component_section = ""
    if builder.is_component():
        col = builder.get_collection()
        col_alias = f"{col.get_api_group()}{col.get_api_version()}"
        col_kind = col.get_api_kind()
        component_section = f"""
// SetCollection sets the collection for a particular workload request.
func (r *{kind}Reconciler) SetCollection(component *{res.import_alias}.{kind}, req *workload.Request) error {{
\tcollection, err := r.GetCollection(component, req)
\tif err != nil || collection == nil {{
\t\treturn fmt.Errorf("unable to set collection, %w", err)
\t}}

\treq.Collection = collection

\treturn r.EnqueueRequestOnCollectionChange(req)
}}

// GetCollection gets a collection for a component given a list.
func (r *{kind}Reconciler) GetCollection(
\tcomponent *{res.import_alias}.{kind},
\treq *workload.Request,
) (*{col_alias}.{col_kind}, error) {{
\tvar collectionList {col_alias}.{col_kind}List

\tif err := r.List(req.Context, &collectionList); err != nil {{
\t\treturn nil, fmt.Errorf("unable to list collection {col_kind}, %w", err)
\t}}

\t// determine if we have requested a specific collection
\tname, namespace := component.Spec.Collection.Name, component.Spec.Collection.Namespace

\tvar collectionRef {res.import_alias}.{kind}CollectionSpec

\thasSpecificCollection := component.Spec.Collection != collectionRef && component.Spec.Collection.Name != ""

\t// if a specific collection has not been requested, we ensure only one exists
\tif !hasSpecificCollection {{
\t\tif len(collectionList.Items) != 1 {{
\t\t\treturn nil, fmt.Errorf("expected only 1 {col_kind} collection, found %v", len(collectionList.Items))
\t\t}}

\t\treturn &collectionList.Items[0], nil
\t}}

\t// find the collection that was requested and return it
\tfor _, collection := range collectionList.Items {{
\t\tif collection.Name == name && collection.Namespace == namespace {{
\t\t\treturn &collection, nil
\t\t}}
\t}}

\treturn nil, workload.ErrCollectionNotFound
}}

// EnqueueRequestOnCollectionChange enqueues a reconcile request when an associated collection object changes.
func (r *{kind}Reconciler) EnqueueRequestOnCollectionChange(req *workload.Request) error {{
\tif len(r.Watches) > 0 {{
\t\tfor _, watched := range r.Watches {{
\t\t\tif reflect.DeepEqual(
\t\t\t\treq.Collection.GetObjectKind().GroupVersionKind(),
\t\t\t\twatched.GetObjectKind().GroupVersionKind(),
\t\t\t) {{
\t\t\t\treturn nil
\t\t\t}}
\t\t}}
\t}}

\t// create a function which maps this specific reconcile request
\tmapFn := func(collection client.Object) []reconcile.Request {{
\t\treturn []reconcile.Request{{
\t\t\t{{
\t\t\t\tNamespacedName: types.NamespacedName{{
\t\t\t\t\tName:      req.Workload.GetName(),
\t\t\t\t\tNamespace: req.Workload.GetNamespace(),
\t\t\t\t}},
\t\t\t}},
\t\t}}
\t}}

\t// watch the collection and use our map function to enqueue the request
\tif err := r.Controller.Watch(
\t\t&source.Kind{{Type: req.Collection}},
\t\thandler.EnqueueRequestsFromMapFunc(mapFn),
\t\tpredicate.Funcs{{
\t\t\tUpdateFunc: func(e event.UpdateEvent) bool {{
\t\t\t\tif !resources.EqualNamespaceName(e.ObjectNew, req.Collection) {{
\t\t\t\t\treturn false
\t\t\t\t}}

\t\t\t\treturn e.ObjectNew != e.ObjectOld
\t\t\t}},
\t\t\tCreateFunc: func(e event.CreateEvent) bool {{
\t\t\t\treturn false
\t\t\t}},
\t\t\tGenericFunc: func(e event.GenericEvent) bool {{
\t\t\t\treturn false
\t\t\t}},
\t\t\tDeleteFunc: func(e event.DeleteEvent) bool {{
\t\t\t\treturn false
\t\t\t}},
\t\t}},
\t); err != nil {{
\t\treturn err
\t}}

\tr.Watches = append(r.Watches, req.Collection)

\treturn nil
}}
"""

    if builder.has_child_resources():
        pkg = builder.get_package_name()
        if builder.is_component():
            convert = (
                f"\tcomponent, collection, err := {pkg}.ConvertWorkload"
                "(req.Workload, req.Collection)"
            )
            gen = f"\tresources, err := {pkg}.Generate(*component, *collection)"
        else:
            convert = (
                f"\tcomponent, err := {pkg}.ConvertWorkload(req.Workload)"
            )
            gen = f"\tresources, err := {pkg}.Generate(*component)"

        get_resources_body = f"""\tresourceObjects := []client.Object{{}}

{convert}
\tif err != nil {{
\t\treturn nil, err
\t}}

\t// create resources in memory
{gen}
\tif err != nil {{
\t\treturn nil, err
\t}}

\t// run through the mutation functions to mutate the resources
\tfor _, resource := range resources {{
\t\tmutatedResources, skip, err := r.Mutate(req, resource)
\t\tif err != nil {{
\t\t\treturn []client.Object{{}}, err
\t\t}}

\t\tif skip {{
\t\t\tcontinue
\t\t}}

\t\tresourceObjects = append(resourceObjects, mutatedResources...)
\t}}

\treturn resourceObjects, nil
"""
    else:
        get_resources_body = "\treturn []client.Object{}, nil\n"

    content = f"""{ctx.boilerplate}

package {res.group}

import (
{_imports_block(ctx, builder)})

// {kind}Reconciler reconciles a {kind} object.
type {kind}Reconciler struct {{
\tclient.Client
\tName         string
\tLog          logr.Logger
\tController   controller.Controller
\tEvents       record.EventRecorder
\tFieldManager string
\tWatches      []client.Object
\tPhases       *phases.Registry
}}

func New{kind}Reconciler(mgr ctrl.Manager) *{kind}Reconciler {{
\treturn &{kind}Reconciler{{
\t\tName:         "{kind}",
\t\tClient:       mgr.GetClient(),
\t\tEvents:       mgr.GetEventRecorderFor("{kind}-Controller"),
\t\tFieldManager: "{kind}-reconciler",
\t\tLog:          ctrl.Log.WithName("controllers").WithName("{res.group}").WithName("{kind}"),
\t\tWatches:      []client.Object{{}},
\t\tPhases:       &phases.Registry{{}},
\t}}
}}

{rbac_markers}
// Until Webhooks are implemented we need to list and watch namespaces to ensure
// they are available before deploying resources,
// See:
//   - https://github.com/vmware-tanzu-labs/operator-builder/issues/141
//   - https://github.com/vmware-tanzu-labs/operator-builder/issues/162

// +kubebuilder:rbac:groups=core,resources=namespaces,verbs=list;watch

// Reconcile is part of the main kubernetes reconciliation loop which aims to
// move the current state of the cluster closer to the desired state.
//
// For more details, check Reconcile and its Result here:
// - https://pkg.go.dev/sigs.k8s.io/controller-runtime@v0.7.2/pkg/reconcile
func (r *{kind}Reconciler) Reconcile(ctx context.Context, request ctrl.Request) (ctrl.Result, error) {{
\treq, err := r.NewRequest(ctx, request)
\tif err != nil {{
{collection_requeue}\t\tif !apierrs.IsNotFound(err) {{
\t\t\treturn ctrl.Result{{}}, err
\t\t}}

\t\treturn ctrl.Result{{}}, nil
\t}}

\tif err := phases.RegisterDeleteHooks(r, req); err != nil {{
\t\treturn ctrl.Result{{}}, err
\t}}

\t// execute the phases
\treturn r.Phases.HandleExecution(r, req)
}}

func (r *{kind}Reconciler) NewRequest(ctx context.Context, request ctrl.Request) (*workload.Request, error) {{
\tcomponent := &{res.import_alias}.{kind}{{}}

\tlog := r.Log.WithValues(
\t\t"kind", component.GetWorkloadGVK().Kind,
\t\t"name", request.Name,
\t\t"namespace", request.Namespace,
\t)

\t// get the component from the cluster
\tif err := r.Get(ctx, request.NamespacedName, component); err != nil {{
\t\tif !apierrs.IsNotFound(err) {{
\t\t\tlog.Error(err, "unable to fetch workload")

\t\t\treturn nil, fmt.Errorf("unable to fetch workload, %w", err)
\t\t}}

\t\treturn nil, err
\t}}

\t// create the workload request
\tworkloadRequest := &workload.Request{{
\t\tContext:  ctx,
\t\tWorkload: component,
\t\tLog:      log,
\t}}

{new_request_return}
}}
{component_section}
// GetResources resources runs the methods to properly construct the resources in memory.
func (r *{kind}Reconciler) GetResources(req *workload.Request) ([]client.Object, error) {{
{get_resources_body}}}

// GetEventRecorder returns the event recorder for writing kubernetes events.
func (r *{kind}Reconciler) GetEventRecorder() record.EventRecorder {{
\treturn r.Events
}}

// GetFieldManager returns the name of the field manager for the controller.
func (r *{kind}Reconciler) GetFieldManager() string {{
\treturn r.FieldManager
}}

// GetLogger returns the logger from the reconciler.
func (r *{kind}Reconciler) GetLogger() logr.Logger {{
\treturn r.Log
}}

// GetName returns the name of the reconciler.
func (r *{kind}Reconciler) GetName() string {{
\treturn r.Name
}}

// GetController returns the controller object associated with the reconciler.
func (r *{kind}Reconciler) GetController() controller.Controller {{
\treturn r.Controller
}}

// GetWatches returns the objects which are current being watched by the reconciler.
func (r *{kind}Reconciler) GetWatches() []client.Object {{
\treturn r.Watches
}}

// SetWatch appends a watch to the list of currently watched objects.
func (r *{kind}Reconciler) SetWatch(watch client.Object) {{
\tr.Watches = append(r.Watches, watch)
}}

// CheckReady will return whether a component is ready.
func (r *{kind}Reconciler) CheckReady(req *workload.Request) (bool, error) {{
\treturn dependencies.{kind}CheckReady(r, req)
}}

// Mutate will run the mutate function for the workload.
func (r *{kind}Reconciler) Mutate(
\treq *workload.Request,
\tobject client.Object,
) ([]client.Object, bool, error) {{
\treturn mutate.{kind}Mutate(r, req, object)
}}

func (r *{kind}Reconciler) SetupWithManager(mgr ctrl.Manager) error {{
\tr.InitializePhases()

\tbaseController, err := ctrl.NewControllerManagedBy(mgr).
\t\tWithEventFilter(predicates.WorkloadPredicates()).
\t\tFor(&{res.import_alias}.{kind}{{}}).
\t\tBuild(r)
\tif err != nil {{
\t\treturn fmt.Errorf("unable to setup controller, %w", err)
\t}}

\tr.Controller = baseController

\treturn nil
}}
"""
    path = os.path.join(
        "controllers", res.group, f"{to_file_name(kind)}_controller.go"
    )
    return File(path, content, IfExists.OVERWRITE)


def phases(ctx: Context) -> File:
    kind = ctx.resource.kind
    content = f"""{ctx.boilerplate}

package {ctx.resource.group}

import (
\t"time"

\t"github.com/nukleros/operator-builder-tools/pkg/controller/phases"
\tctrl "sigs.k8s.io/controller-runtime"
)

// InitializePhases defines what phases should be run for each event loop. phases are executed
// in the order they are listed.
func (r *{kind}Reconciler) InitializePhases() {{
\t// Create Phases
\tr.Phases.Register(
\t\t"Dependency",
\t\tphases.DependencyPhase,
\t\tphases.CreateEvent,
\t\tphases.WithCustomRequeueResult(ctrl.Result{{RequeueAfter: 5 * time.Second }}),
\t)

\tr.Phases.Register(
\t\t"Create-Resources",
\t\tphases.CreateResourcesPhase,
\t\tphases.CreateEvent,
\t)

\tr.Phases.Register(
\t\t"Check-Ready",
\t\tphases.CheckReadyPhase,
\t\tphases.CreateEvent,
\t\tphases.WithCustomRequeueResult(ctrl.Result{{RequeueAfter: 5 * time.Second }}),
\t)

\tr.Phases.Register(
\t\t"Complete",
\t\tphases.CompletePhase,
\t\tphases.CreateEvent,
\t)

\t// Update Phases
\tr.Phases.Register(
\t\t"Dependency",
\t\tphases.DependencyPhase,
\t\tphases.UpdateEvent,
\t\tphases.WithCustomRequeueResult(ctrl.Result{{RequeueAfter: 5 * time.Second }}),
\t)

\tr.Phases.Register(
\t\t"Create-Resources",
\t\tphases.CreateResourcesPhase,
\t\tphases.UpdateEvent,
\t)

\tr.Phases.Register(
\t\t"Check-Ready",
\t\tphases.CheckReadyPhase,
\t\tphases.UpdateEvent,
\t\tphases.WithCustomRequeueResult(ctrl.Result{{RequeueAfter: 5 * time.Second }}),
\t)

\tr.Phases.Register(
\t\t"Complete",
\t\tphases.CompletePhase,
\t\tphases.UpdateEvent,
\t)

\t// Delete Phases
\tr.Phases.Register(
\t\t"DeletionComplete",
\t\tphases.DeletionCompletePhase,
\t\tphases.DeleteEvent,
\t)
}}
"""
    path = os.path.join(
        "controllers",
        ctx.resource.group,
        f"{to_file_name(kind)}_phases.go",
    )
    return File(path, content, IfExists.SKIP)


def suite_test(ctx: Context) -> tuple[File, Fragments]:
    """controllers/<group>/suite_test.go (reference
    controller_suitetest.go; scaffolded by the upstream golang/v3 plugin
    in the reference's bundle, re-implemented here)."""
    res = ctx.resource

    crd_path = '"..", ".."' if ctx.multi_group and res.group else '".."'
    package = res.group if ctx.multi_group and res.group else "controllers"

    content = f"""{ctx.boilerplate}

package {package}

import (
\t"path/filepath"
\t"testing"
\t. "github.com/onsi/ginkgo"
\t. "github.com/onsi/gomega"
\t"k8s.io/client-go/kubernetes/scheme"
\t"k8s.io/client-go/rest"
\t"sigs.k8s.io/controller-runtime/pkg/client"
\t"sigs.k8s.io/controller-runtime/pkg/envtest"
\t"sigs.k8s.io/controller-runtime/pkg/envtest/printer"
\tlogf "sigs.k8s.io/controller-runtime/pkg/log"
\t"sigs.k8s.io/controller-runtime/pkg/log/zap"
\t{SUITE_IMPORT_MARKER}
)

// These tests use Ginkgo (BDD-style Go testing framework). Refer to
// http://onsi.github.io/ginkgo/ to learn more about Ginkgo.

var cfg *rest.Config
var k8sClient client.Client
var testEnv *envtest.Environment

func TestAPIs(t *testing.T) {{
\tRegisterFailHandler(Fail)

\tRunSpecsWithDefaultAndCustomReporters(t,
\t"Controller Suite",
\t[]Reporter{{printer.NewlineReporter{{}}}})
}}

var _ = BeforeSuite(func() {{
\tlogf.SetLogger(zap.New(zap.WriteTo(GinkgoWriter), zap.UseDevMode(true)))

\tBy("bootstrapping test environment")
\ttestEnv = &envtest.Environment{{
\t\tCRDDirectoryPaths:     []string{{filepath.Join({crd_path}, "config", "crd", "bases")}},
\t\tErrorIfCRDPathMissing: true,
\t}}

\tcfg, err := testEnv.Start()
\tExpect(err).NotTo(HaveOccurred())
\tExpect(cfg).NotTo(BeNil())

\t{SUITE_SCHEME_MARKER}

\tk8sClient, err = client.New(cfg, client.Options{{Scheme: scheme.Scheme}})
\tExpect(err).NotTo(HaveOccurred())
\tExpect(k8sClient).NotTo(BeNil())

}}, 60)

var _ = AfterSuite(func() {{
\tBy("tearing down the test environment")
\terr := testEnv.Stop()
\tExpect(err).NotTo(HaveOccurred())
}})
"""
    path = os.path.join("controllers", res.group, "suite_test.go")

    file = File(path, content, IfExists.SKIP)
    fragments = Fragments(
        path=path,
        fragments={
            SUITE_IMPORT_MARKER: [
                f'{res.import_alias} "{res.path}"\n'
            ],
            SUITE_SCHEME_MARKER: [
                f"err = {res.import_alias}.AddToScheme(scheme.Scheme)\n"
                "Expect(err).NotTo(HaveOccurred())\n\n"
            ],
        },
    )
    return file, fragments
