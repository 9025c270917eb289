"""E2E test templates for the generated operator.

Parity targets: reference templates/test/e2e/e2e.go:22-875 (the common
harness) and templates/test/e2e/workloads.go:44-210 (per-workload tests).
Feature surface reproduced: build-tag gating
(``e2e_test``), a serial collection suite followed by a parallel component
suite, per-test namespaces, create + wait-for-children, delete a
whitelisted child and wait for reconciliation to restore it, controller
log error scan, and DEPLOY / DEPLOY_IN_CLUSTER / TEARDOWN env switches.
The harness here is an original implementation of that surface; the
symbols shared with the per-workload test files (E2ETest,
E2EComponentTestSuite, E2ECollectionTestSuite, test* helpers) keep the
same names because the generated files reference each other.
"""

from __future__ import annotations

from ...utils import to_file_name
from ...workload.kinds import Workload
from ..context import Context
from ..machinery import File, IfExists

E2E_HARNESS = """//go:build e2e_test
// +build e2e_test

__BOILERPLATE__

package e2e_test

import (
\t"bytes"
\t"context"
\t"fmt"
\t"io"
\t"os"
\t"os/exec"
\t"strings"
\t"testing"
\t"time"

\t"github.com/stretchr/testify/require"
\t"github.com/stretchr/testify/suite"
\t"gopkg.in/yaml.v2"

\tappsv1 "k8s.io/api/apps/v1"
\tv1 "k8s.io/api/core/v1"
\t"sigs.k8s.io/controller-runtime/pkg/client"
\tk8syaml "sigs.k8s.io/yaml"

\t"k8s.io/apimachinery/pkg/api/errors"
\tmetav1 "k8s.io/apimachinery/pkg/apis/meta/v1"
\t"k8s.io/apimachinery/pkg/apis/meta/v1/unstructured"
\t"k8s.io/apimachinery/pkg/labels"
\t"k8s.io/apimachinery/pkg/runtime/schema"
\tserializer "k8s.io/apimachinery/pkg/runtime/serializer/yaml"

\t"k8s.io/client-go/dynamic"
\t"k8s.io/client-go/kubernetes"
\t"k8s.io/client-go/rest"
\t"k8s.io/client-go/tools/clientcmd"

\t"github.com/nukleros/operator-builder-tools/pkg/controller/workload"
\t"github.com/nukleros/operator-builder-tools/pkg/resources"
\tkbresource "sigs.k8s.io/kubebuilder/v3/pkg/model/resource"
)

// timing and identity knobs for the suite.
const (
\tcontrollerName          = "controller-manager"
\tcontrollerKustomization = "../../config/default/kustomization.yaml"
\twaitTimeout             = 90 * time.Second
\twaitInterval            = 3 * time.Second
)

// kinds we are allowed to delete when testing that the controller
// restores deleted children.
var deletableWhitelist = []string{
\t"Deployment",
\t"Secret",
\t"ConfigMap",
\t"DaemonSet",
\t"Pod",
\t"Service",
\t"Ingress",
\t"StorageClass",
}

// E2ETestSuiteConfig carries the clients and controller metadata shared
// by every test in the suite.
type E2ETestSuiteConfig struct {
\tdynamicClient    dynamic.Interface
\tclient           kubernetes.Clientset
\tcontrollerConfig controllerConfig
\ttests            []*E2ETest
}

type controllerConfig struct {
\tNamespace string `yaml:"namespace"`
\tPrefix    string `yaml:"namePrefix"`
}

// E2ECollectionTestSuite runs collection workload tests serially, before
// any component tests.
type E2ECollectionTestSuite struct {
\tsuite.Suite

\tsuiteConfig E2ETestSuiteConfig
}

// E2EComponentTestSuite runs component/standalone workload tests in
// parallel, after collections exist.
type E2EComponentTestSuite struct {
\tsuite.Suite

\tsuiteConfig E2ETestSuiteConfig
}

// E2ETest is the per-workload test harness.
type E2ETest struct {
\tsuiteConfig        *E2ETestSuiteConfig
\tnamespace          string
\tsampleManifestFile string
\tunstructured       *unstructured.Unstructured
\tworkload           workload.Workload
\tcollectionTester   *E2ETest
\tchildren           []client.Object
\tgetChildrenFunc    getChildren
\tlogSyntax          string
}

type getChildren func(*E2ETest) error
type readyChecker func() (bool, error)

// TestMain drives the whole suite: collections first (serial), then
// components (parallel), then teardown and a controller log scan.
func TestMain(t *testing.T) {
\te2eTestSuite := new(E2ETestSuiteConfig)
\trequire.NoErrorf(t, setupSuite(e2eTestSuite), "error setting up test suite")

\tcollectionSuite := &E2ECollectionTestSuite{suiteConfig: *e2eTestSuite}
\tcomponentSuite := &E2EComponentTestSuite{suiteConfig: *e2eTestSuite}

\tt.Run("TestE2ESuite", func(t *testing.T) {
\t\tsuite.Run(t, collectionSuite)
\t\tsuite.Run(t, componentSuite)
\t})

\tcomponentSuite.teardown()
\tcollectionSuite.teardown()

\tif os.Getenv("DEPLOY_IN_CLUSTER") == "true" {
\t\trequire.NoErrorf(t, testControllerLogsNoErrors(e2eTestSuite, ""), "found errors in controller logs")
\t}

\trequire.NoErrorf(t, finalTeardown(), "error tearing down test suite")
}

// setupSuite builds the clients from KUBECONFIG (or ~/.kube/config),
// loads the controller kustomization metadata, and deploys if requested.
func setupSuite(s *E2ETestSuiteConfig) error {
\tvar err error

\tvar config *rest.Config

\tif os.Getenv("KUBECONFIG") != "" {
\t\tconfig, err = clientcmd.BuildConfigFromFlags("", os.Getenv("KUBECONFIG"))
\t} else {
\t\tconfig, err = clientcmd.BuildConfigFromFlags("", os.Getenv("HOME")+"/.kube/config")
\t}

\tif err != nil {
\t\treturn fmt.Errorf("unable to create rest config from kubeconfig; %w", err)
\t}

\trestClient, err := kubernetes.NewForConfig(config)
\tif err != nil {
\t\treturn fmt.Errorf("unable create rest client from kubeconfig; %w", err)
\t}

\ts.client = *restClient

\ts.dynamicClient, err = dynamic.NewForConfig(config)
\tif err != nil {
\t\treturn fmt.Errorf("unable to create dynamic client from kubeconfig; %w", err)
\t}

\tif err := readYamlFile(controllerKustomization, &s.controllerConfig); err != nil {
\t\treturn fmt.Errorf("unable to fetch controller configuration; %w", err)
\t}

\treturn deploy(s)
}

// SetupTest marks every component test parallel.
func (s *E2EComponentTestSuite) SetupTest() {
\ts.T().Parallel()
}

// setup loads the sample manifest, aligns namespaces, resolves children
// and creates the per-test namespace.
func (tester *E2ETest) setup() error {
\tyamlFile, err := readYamlManifest(tester.sampleManifestFile, tester.unstructured)
\tif err != nil {
\t\treturn fmt.Errorf("unable to fetch sample manifest; %w", err)
\t}

\tif err := k8syaml.Unmarshal(yamlFile, tester.workload); err != nil {
\t\treturn fmt.Errorf("unable to unmarshal yaml to api object; %w", err)
\t}

\ttester.unstructured.SetNamespace(tester.namespace)
\ttester.workload.SetNamespace(tester.namespace)

\tif tester.collectionTester != nil {
\t\tcollection := &unstructured.Unstructured{}

\t\tcollectionYaml, err := readYamlManifest(tester.collectionTester.sampleManifestFile, collection)
\t\tif err != nil {
\t\t\treturn fmt.Errorf("unable to fetch sample collection manifest; %w", err)
\t\t}

\t\tif err := k8syaml.Unmarshal(collectionYaml, tester.collectionTester.workload); err != nil {
\t\t\treturn fmt.Errorf("unable to unmarshal collection yaml to api object; %w", err)
\t\t}

\t\ttester.collectionTester.unstructured.SetNamespace(tester.collectionTester.namespace)
\t\ttester.collectionTester.workload.SetNamespace(tester.collectionTester.namespace)
\t}

\tif err := tester.getChildrenFunc(tester); err != nil {
\t\treturn fmt.Errorf("unable to resolve child resources; %w", err)
\t}

\t// cluster-scoped testers have no namespace and skip namespace creation
\tif tester.namespace != "" {
\t\tif err := createNamespaceForTest(tester); err != nil {
\t\t\treturn fmt.Errorf("failed to create namespace for test; %w", err)
\t\t}
\t}

\treturn nil
}

// deploy optionally installs CRDs and the controller:
//   DEPLOY="true"            -> make install (+ docker-build/push + deploy)
//   DEPLOY_IN_CLUSTER="true" -> wait for the in-cluster controller
func deploy(s *E2ETestSuiteConfig) error {
\tif os.Getenv("DEPLOY") == "true" {
\t\tif _, err := exec.Command("make", "-C", "../..", "install").Output(); err != nil {
\t\t\treturn fmt.Errorf("failed to run 'make install' target; %w", err)
\t\t}
\t}

\tif os.Getenv("DEPLOY_IN_CLUSTER") == "true" {
\t\tif os.Getenv("DEPLOY") == "true" {
\t\t\tfor _, target := range []string{"docker-build", "docker-push", "deploy"} {
\t\t\t\tif _, err := exec.Command("make", "-C", "../..", target).Output(); err != nil {
\t\t\t\t\treturn fmt.Errorf("failed to run 'make %s' target; %w", target, err)
\t\t\t\t}
\t\t\t}
\t\t}

\t\tif err := waitForController(s); err != nil {
\t\t\treturn fmt.Errorf("failed to wait for controller for test; %w", err)
\t\t}
\t}

\treturn nil
}

// finalTeardown undeploys/uninstalls when TEARDOWN="true".
func finalTeardown() error {
\tif os.Getenv("TEARDOWN") == "true" {
\t\ttarget := "uninstall"
\t\tif os.Getenv("DEPLOY_IN_CLUSTER") == "true" {
\t\t\ttarget = "undeploy"
\t\t}

\t\tif _, err := exec.Command("make", "-C", "../..", target).Output(); err != nil {
\t\t\treturn fmt.Errorf("failed to run 'make %s' target with error; %w", target, err)
\t\t}
\t}

\treturn nil
}

func teardownSuite(s *E2ETestSuiteConfig) error {
\tfor _, e2eTest := range s.tests {
\t\tif err := deleteCustomResource(e2eTest); err != nil {
\t\t\treturn fmt.Errorf("failed to delete custom resource: %+v; %w", e2eTest, err)
\t\t}

\t\tif e2eTest.namespace != "" {
\t\t\tif err := deleteNamespaceForTest(e2eTest); err != nil {
\t\t\t\treturn fmt.Errorf("failed to delete namespace during teardown: %s; %w", e2eTest.namespace, err)
\t\t\t}
\t\t}
\t}

\treturn nil
}

func (s *E2ECollectionTestSuite) teardown() {
\tif len(s.suiteConfig.tests) > 0 {
\t\trequire.NoErrorf(s.T(), teardownSuite(&s.suiteConfig), "unable to teardown collection test suite")
\t}
}

func (s *E2EComponentTestSuite) teardown() {
\tif len(s.suiteConfig.tests) > 0 {
\t\trequire.NoErrorf(s.T(), teardownSuite(&s.suiteConfig), "unable to teardown component test suite")
\t}
}

//
// manifest + client helpers
//

func readYamlManifest(path string, destination *unstructured.Unstructured) ([]byte, error) {
\tyamlFile, err := os.ReadFile(path)
\tif err != nil {
\t\treturn nil, fmt.Errorf("unable to read file %s; %w", path, err)
\t}

\tdec := serializer.NewDecodingSerializer(unstructured.UnstructuredJSONScheme)

\tif _, _, err = dec.Decode(yamlFile, nil, destination); err != nil {
\t\treturn nil, fmt.Errorf("error decoding sample manifest %s; %w\\n\\nwith data: %s", path, err, yamlFile)
\t}

\treturn yamlFile, nil
}

func readYamlFile(path string, destination interface{}) error {
\tyamlFile, err := os.ReadFile(path)
\tif err != nil {
\t\treturn fmt.Errorf("unable to read file %s; %w", path, err)
\t}

\tif err = yaml.Unmarshal(yamlFile, destination); err != nil {
\t\treturn fmt.Errorf("unable to unmarshal yaml file %s; %w", path, err)
\t}

\treturn nil
}

func namespaceExists(tester *E2ETest) (bool, error) {
\t_, err := tester.suiteConfig.client.CoreV1().Namespaces().Get(
\t\tcontext.TODO(),
\t\ttester.namespace,
\t\tmetav1.GetOptions{},
\t)
\tif err != nil {
\t\tif errors.IsNotFound(err) {
\t\t\treturn false, nil
\t\t}

\t\treturn false, err
\t}

\treturn true, nil
}

func getPlural(kind string) string {
\tpluralMap := map[string]string{
\t\t"resourcequota": "resourcequotas",
\t}

\tplural := kbresource.RegularPlural(kind)

\tif pluralMap[plural] != "" {
\t\treturn pluralMap[plural]
\t}

\treturn plural
}

func getDeletableChild(tester *E2ETest) client.Object {
\tfor _, whitelistKind := range deletableWhitelist {
\t\tfor _, child := range tester.children {
\t\t\tif child.GetObjectKind().GroupVersionKind().Kind == whitelistKind {
\t\t\t\treturn child
\t\t\t}
\t\t}
\t}

\treturn nil
}

func getResourceGVR(resource client.Object) schema.GroupVersionResource {
\treturn schema.GroupVersionResource{
\t\tGroup:    resource.GetObjectKind().GroupVersionKind().Group,
\t\tVersion:  resource.GetObjectKind().GroupVersionKind().Version,
\t\tResource: getPlural(strings.ToLower(resource.GetObjectKind().GroupVersionKind().Kind)),
\t}
}

func getClientForResource(tester *E2ETest, resource client.Object) dynamic.ResourceInterface {
\tif tester.namespace != "" {
\t\treturn tester.suiteConfig.dynamicClient.Resource(getResourceGVR(resource)).
\t\t\tNamespace(tester.namespace)
\t}

\treturn tester.suiteConfig.dynamicClient.Resource(getResourceGVR(resource)).
\t\tNamespace(resource.GetNamespace())
}

func getControllerDeployment(s *E2ETestSuiteConfig) (*appsv1.Deployment, error) {
\treturn s.client.
\t\tAppsV1().Deployments(s.controllerConfig.Namespace).
\t\tGet(context.TODO(), (s.controllerConfig.Prefix + controllerName), metav1.GetOptions{})
}

func createNamespaceForTest(tester *E2ETest) error {
\texists, err := namespaceExists(tester)
\tif exists || err != nil {
\t\treturn err
\t}

\tnamespace := &v1.Namespace{
\t\tTypeMeta: metav1.TypeMeta{
\t\t\tAPIVersion: resources.NamespaceVersion,
\t\t\tKind:       resources.NamespaceKind,
\t\t},
\t\tObjectMeta: metav1.ObjectMeta{
\t\t\tName: tester.namespace,
\t\t},
\t}

\t_, err = tester.suiteConfig.client.
\t\tCoreV1().Namespaces().
\t\tCreate(context.TODO(), namespace, metav1.CreateOptions{})

\treturn err
}

func getControllerLogs(s *E2ETestSuiteConfig) (string, error) {
\tdeployment, err := getControllerDeployment(s)
\tif err != nil {
\t\treturn "", fmt.Errorf("unable to retrieve controller deployment; %w", err)
\t}

\tpodListOpts := metav1.ListOptions{
\t\tLabelSelector: labels.SelectorFromSet(deployment.Spec.Template.Labels).String(),
\t}

\tcontrollerPods, err := s.client.CoreV1().Pods(s.controllerConfig.Namespace).List(context.TODO(), podListOpts)
\tif err != nil {
\t\treturn "", fmt.Errorf("unable to retrieve controller pods; %w", err)
\t}

\tbuf := new(bytes.Buffer)

\tfor _, pod := range controllerPods.Items {
\t\tfor _, container := range pod.Spec.Containers {
\t\t\tpodLogOpts := v1.PodLogOptions{Container: container.Name}
\t\t\treq := s.client.CoreV1().Pods(pod.Namespace).GetLogs(pod.Name, &podLogOpts)

\t\t\tpodLogs, err := req.Stream(context.TODO())
\t\t\tif err != nil {
\t\t\t\treturn "", fmt.Errorf("error opening log stream for pod %s/%s; %w", pod.Namespace, pod.Name, err)
\t\t\t}

\t\t\tdefer podLogs.Close()

\t\t\tif _, err := io.Copy(buf, podLogs); err != nil {
\t\t\t\treturn "", fmt.Errorf("error storing logs to string buffer; %w", err)
\t\t\t}
\t\t}
\t}

\treturn buf.String(), nil
}

func updateResource(tester *E2ETest, resource client.Object) error {
\tunstructuredResource, err := resources.ToUnstructured(resource)
\tif err != nil {
\t\treturn err
\t}

\t_, err = getClientForResource(tester, resource).
\t\tUpdate(context.TODO(), unstructuredResource, metav1.UpdateOptions{})

\treturn err
}

func deleteResource(tester *E2ETest, resource client.Object) error {
\treturn getClientForResource(tester, resource).
\t\tDelete(context.TODO(), resource.GetName(), metav1.DeleteOptions{})
}

func deleteCustomResource(tester *E2ETest) error {
\tcrClient := getClientForResource(tester, tester.unstructured)

\t_, err := crClient.Get(context.TODO(), tester.unstructured.GetName(), metav1.GetOptions{})
\tif err != nil {
\t\tif errors.IsNotFound(err) {
\t\t\treturn nil
\t\t}

\t\treturn err
\t}

\tif err := crClient.Delete(context.TODO(), tester.unstructured.GetName(), metav1.DeleteOptions{}); err != nil {
\t\treturn fmt.Errorf("error deleting custom resource: %+v; %w", tester.unstructured, err)
\t}

\treturn waitForMissingResources(tester)
}

func deleteNamespaceForTest(tester *E2ETest) error {
\terr := tester.suiteConfig.client.
\t\tCoreV1().Namespaces().
\t\tDelete(context.TODO(), tester.namespace, metav1.DeleteOptions{})
\tif err != nil {
\t\treturn err
\t}

\tnamespaceIsMissing := func() (bool, error) {
\t\texists, err := namespaceExists(tester)
\t\tif err != nil {
\t\t\treturn false, err
\t\t}

\t\treturn !exists, nil
\t}

\treturn waitFor(namespaceIsMissing)
}

//
// wait helpers
//

func waitForMissingResources(tester *E2ETest) error {
\tchildResourcesAreMissing := func() (bool, error) {
\t\tfor _, child := range tester.children {
\t\t\t_, err := getClientForResource(tester, child).
\t\t\t\tGet(context.TODO(), child.GetName(), metav1.GetOptions{})

\t\t\tif err == nil {
\t\t\t\treturn false, nil
\t\t\t}

\t\t\tif errors.IsNotFound(err) {
\t\t\t\tcontinue
\t\t\t}

\t\t\treturn false, err
\t\t}

\t\treturn true, nil
\t}

\treturn waitFor(childResourcesAreMissing)
}

func waitForChildResources(tester *E2ETest) error {
\tchildResourcesAreReady := func() (bool, error) {
\t\tchildResourceClusterObjects := make([]client.Object, len(tester.children))

\t\tfor i, child := range tester.children {
\t\t\tchildResourceClusterObject, err := getClientForResource(tester, child).
\t\t\t\tGet(context.TODO(), child.GetName(), metav1.GetOptions{})
\t\t\tif err != nil {
\t\t\t\treturn false, fmt.Errorf("unable to get child resource from cluster: %+v; %w", child, err)
\t\t\t}

\t\t\tchildResourceClusterObjects[i] = childResourceClusterObject
\t\t}

\t\treturn resources.AreReady(childResourceClusterObjects...)
\t}

\treturn waitFor(childResourcesAreReady)
}

func waitForCustomResource(tester *E2ETest) error {
\tcustomResourceIsReady := func() (bool, error) {
\t\tcustomResource, err := getClientForResource(tester, tester.unstructured).
\t\t\tGet(context.TODO(), tester.unstructured.GetName(), metav1.GetOptions{})
\t\tif err != nil {
\t\t\treturn false, fmt.Errorf("unable to get custom resource from cluster: %+v; %w", customResource, err)
\t\t}

\t\tif customResource.Object["status"] == nil {
\t\t\treturn false, nil
\t\t}

\t\tcreateStatus := customResource.Object["status"].(map[string]interface{})["created"]
\t\tif createStatus != nil {
\t\t\tcreated, ok := createStatus.(bool)
\t\t\tif !ok {
\t\t\t\treturn false, fmt.Errorf("unable to determine custom resource status")
\t\t\t}

\t\t\treturn created, nil
\t\t}

\t\treturn false, nil
\t}

\treturn waitFor(customResourceIsReady)
}

func waitForController(s *E2ETestSuiteConfig) error {
\tdeploymentIsReady := func() (bool, error) {
\t\tdeployment, err := getControllerDeployment(s)
\t\tif err != nil {
\t\t\treturn false, err
\t\t}

\t\treturn resources.IsReady(deployment)
\t}

\treturn waitFor(deploymentIsReady)
}

func waitFor(isReady readyChecker) error {
\ttimeout, interval := time.After(waitTimeout), time.Tick(waitInterval)

\tfor {
\t\tselect {
\t\tcase <-timeout:
\t\t\treturn fmt.Errorf("timed out waiting for resource")
\t\tcase <-interval:
\t\t\tready, err := isReady()
\t\t\tif err != nil {
\t\t\t\treturn fmt.Errorf("error waiting for resource to be ready, %w", err)
\t\t\t}

\t\t\tif ready {
\t\t\t\treturn nil
\t\t\t}
\t\t}
\t}
}

//
// test bodies shared by every workload test file
//

func testCreateCustomResource(tester *E2ETest) error {
\t_, err := getClientForResource(tester, tester.unstructured).
\t\tCreate(context.TODO(), tester.unstructured, metav1.CreateOptions{})
\tif err != nil {
\t\treturn fmt.Errorf("error creating custom resource: %+v; %w", tester.unstructured, err)
\t}

\tif err = waitForCustomResource(tester); err != nil {
\t\treturn fmt.Errorf("failed waiting for custom resource ready status: %v; %w", tester.unstructured, err)
\t}

\tif err = waitForChildResources(tester); err != nil {
\t\treturn fmt.Errorf("child resources are not in a ready state: %v; %w", tester.unstructured, err)
\t}

\treturn nil
}

func testDeleteChildResource(tester *E2ETest) error {
\tchildToDelete := getDeletableChild(tester)
\tif childToDelete != nil {
\t\tif err := deleteResource(tester, childToDelete); err != nil {
\t\t\treturn fmt.Errorf("failed deleting child resource;: %+v; %w", childToDelete, err)
\t\t}

\t\tif err := waitForChildResources(tester); err != nil {
\t\t\treturn fmt.Errorf(
\t\t\t\t"failed waiting for reconciliation after child deletion for resource: %+v; %w",
\t\t\t\tchildToDelete,
\t\t\t\terr,
\t\t\t)
\t\t}
\t}

\treturn nil
}

func testControllerLogsNoErrors(s *E2ETestSuiteConfig, searchSyntax string) error {
\tlogs, err := getControllerLogs(s)
\tif err != nil {
\t\treturn fmt.Errorf("failed fetching controller logs; %w", err)
\t}

\tfoundErrors := []string{}

\tfor _, logLine := range strings.Split(logs, "\\n") {
\t\tif strings.Contains(logLine, "ERROR") && strings.Contains(logLine, searchSyntax) {
\t\t\tfoundErrors = append(foundErrors, logLine)
\t\t}
\t}

\tif len(foundErrors) > 0 {
\t\treturn fmt.Errorf("found errors in controller: +%v", foundErrors)
\t}

\treturn nil
}
"""


def e2e_test(ctx: Context) -> File:
    return File(
        "test/e2e/e2e_test.go",
        E2E_HARNESS.replace("__BOILERPLATE__", ctx.boilerplate),
        IfExists.OVERWRITE,
    )


def workload_test(ctx: Context, builder: Workload) -> File:
    res = ctx.resource
    kind = res.kind
    tester = f"{res.import_alias}{kind}"

    # the per-test namespace (cluster-scoped testers have none)
    namespace = ""
    if not builder.is_cluster_scoped():
        namespace = "-".join(
            [
                "test",
                builder.get_api_group().lower(),
                builder.get_api_version().lower(),
                builder.get_api_kind().lower(),
            ]
        )

    sample_path = "../../config/samples/{}_{}_{}.yaml".format(
        res.group, res.version, to_file_name(kind)
    )

    if builder.is_component():
        col = builder.get_collection()
        col_tester = (
            f"{col.get_api_group().lower()}{col.get_api_version().lower()}"
            f"{col.get_api_kind()}"
        )
        col_namespace = ""
        if not col.is_cluster_scoped():
            col_namespace = "-".join(
                [
                    "test",
                    col.get_api_group().lower(),
                    col.get_api_version().lower(),
                    col.get_api_kind().lower(),
                ]
            )
        collection_field = (
            f'\t\tcollectionTester:   {col_tester}NewHarness('
            f'"{col_namespace}"),\n'
        )
        convert_call = (
            f"\tworkload, collection, err := "
            f"{builder.get_package_name()}.ConvertWorkload(tester.workload, "
            "tester.collectionTester.workload)"
        )
        generate_call = (
            f"\tresourceObjects, err := "
            f"{builder.get_package_name()}.Generate(*workload, *collection)"
        )
    else:
        collection_field = ""
        convert_call = (
            f"\tworkload, err := "
            f"{builder.get_package_name()}.ConvertWorkload(tester.workload)"
        )
        generate_call = (
            f"\tresourceObjects, err := "
            f"{builder.get_package_name()}.Generate(*workload)"
        )

    suite_type = (
        "E2ECollectionTestSuite"
        if builder.is_collection()
        else "E2EComponentTestSuite"
    )

    multi_test = ""
    if not builder.is_cluster_scoped() and not builder.is_collection():
        multi_test = f"""
func (testSuite *E2EComponentTestSuite) Test_{tester}Multi() {{
\ttester := {tester}NewHarness("{namespace}-2")
\ttester.{tester}Test(testSuite)
}}
"""

    content = f"""//go:build e2e_test
// +build e2e_test

{ctx.boilerplate}

package e2e_test

import (
\t"fmt"
\t"os"

\t"github.com/stretchr/testify/require"

\t"k8s.io/apimachinery/pkg/apis/meta/v1/unstructured"

\t{res.import_alias} "{res.path}"
\t"{res.path}/{builder.get_package_name()}"
)

//
// {tester} tests
//
func {tester}ChildrenFuncs(tester *E2ETest) error {{
\tif len({builder.get_package_name()}.CreateFuncs) == 0 {{
\t\treturn nil
\t}}

{convert_call}
\tif err != nil {{
\t\treturn fmt.Errorf("error in workload conversion; %w", err)
\t}}

{generate_call}
\tif err != nil {{
\t\treturn fmt.Errorf("unable to create objects in memory; %w", err)
\t}}

\ttester.children = resourceObjects

\treturn nil
}}

func {tester}NewHarness(namespace string) *E2ETest {{
\treturn &E2ETest{{
\t\tnamespace:          namespace,
\t\tunstructured:       &unstructured.Unstructured{{}},
\t\tworkload:           &{res.import_alias}.{kind}{{}},
\t\tsampleManifestFile: "{sample_path}",
\t\tgetChildrenFunc:    {tester}ChildrenFuncs,
\t\tlogSyntax:          "controllers.{res.group}.{kind}",
{collection_field}\t}}
}}

func (tester *E2ETest) {tester}Test(testSuite *{suite_type}) {{
\ttestSuite.suiteConfig.tests = append(testSuite.suiteConfig.tests, tester)
\ttester.suiteConfig = &testSuite.suiteConfig
\trequire.NoErrorf(testSuite.T(), tester.setup(), "failed to setup test")

\t// create the custom resource and wait for its children to be ready
\trequire.NoErrorf(testSuite.T(), testCreateCustomResource(tester), "failed to create custom resource")

\t// delete a whitelisted child and wait for the controller to restore it
\trequire.NoErrorf(testSuite.T(), testDeleteChildResource(tester), "failed to reconcile deletion of a child resource")

\t// verify the controller logged no errors for this workload
\tif os.Getenv("DEPLOY_IN_CLUSTER") == "true" {{
\t\trequire.NoErrorf(testSuite.T(), testControllerLogsNoErrors(tester.suiteConfig, tester.logSyntax), "found errors in controller logs")
\t}}
}}

func (testSuite *{suite_type}) Test_{tester}() {{
\ttester := {tester}NewHarness("{namespace}")
\ttester.{tester}Test(testSuite)
}}
{multi_test}"""

    path = "test/e2e/{}_{}_{}_test.go".format(
        res.group, res.version, kind.lower()
    )
    return File(path, content, IfExists.SKIP)
