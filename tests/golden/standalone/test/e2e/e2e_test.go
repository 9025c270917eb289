//go:build e2e_test
// +build e2e_test

/*
Copyright 2021.

Licensed under the Apache License, Version 2.0 (the "License");
you may not use this file except in compliance with the License.
You may obtain a copy of the License at

    http://www.apache.org/licenses/LICENSE-2.0

Unless required by applicable law or agreed to in writing, software
distributed under the License is distributed on an "AS IS" BASIS,
WITHOUT WARRANTIES OR CONDITIONS OF ANY KIND, either express or implied.
See the License for the specific language governing permissions and
limitations under the License.
*/

package e2e_test

import (
	"bytes"
	"context"
	"fmt"
	"io"
	"os"
	"os/exec"
	"strings"
	"testing"
	"time"

	"github.com/stretchr/testify/require"
	"github.com/stretchr/testify/suite"
	"gopkg.in/yaml.v2"

	appsv1 "k8s.io/api/apps/v1"
	v1 "k8s.io/api/core/v1"
	"sigs.k8s.io/controller-runtime/pkg/client"
	k8syaml "sigs.k8s.io/yaml"

	"k8s.io/apimachinery/pkg/api/errors"
	metav1 "k8s.io/apimachinery/pkg/apis/meta/v1"
	"k8s.io/apimachinery/pkg/apis/meta/v1/unstructured"
	"k8s.io/apimachinery/pkg/labels"
	"k8s.io/apimachinery/pkg/runtime/schema"
	serializer "k8s.io/apimachinery/pkg/runtime/serializer/yaml"

	"k8s.io/client-go/dynamic"
	"k8s.io/client-go/kubernetes"
	"k8s.io/client-go/rest"
	"k8s.io/client-go/tools/clientcmd"

	"github.com/nukleros/operator-builder-tools/pkg/controller/workload"
	"github.com/nukleros/operator-builder-tools/pkg/resources"
	kbresource "sigs.k8s.io/kubebuilder/v3/pkg/model/resource"
)

// timing and identity knobs for the suite.
const (
	controllerName          = "controller-manager"
	controllerKustomization = "../../config/default/kustomization.yaml"
	waitTimeout             = 90 * time.Second
	waitInterval            = 3 * time.Second
)

// kinds we are allowed to delete when testing that the controller
// restores deleted children.
var deletableWhitelist = []string{
	"Deployment",
	"Secret",
	"ConfigMap",
	"DaemonSet",
	"Pod",
	"Service",
	"Ingress",
	"StorageClass",
}

// E2ETestSuiteConfig carries the clients and controller metadata shared
// by every test in the suite.
type E2ETestSuiteConfig struct {
	dynamicClient    dynamic.Interface
	client           kubernetes.Clientset
	controllerConfig controllerConfig
	tests            []*E2ETest
}

type controllerConfig struct {
	Namespace string `yaml:"namespace"`
	Prefix    string `yaml:"namePrefix"`
}

// E2ECollectionTestSuite runs collection workload tests serially, before
// any component tests.
type E2ECollectionTestSuite struct {
	suite.Suite

	suiteConfig E2ETestSuiteConfig
}

// E2EComponentTestSuite runs component/standalone workload tests in
// parallel, after collections exist.
type E2EComponentTestSuite struct {
	suite.Suite

	suiteConfig E2ETestSuiteConfig
}

// E2ETest is the per-workload test harness.
type E2ETest struct {
	suiteConfig        *E2ETestSuiteConfig
	namespace          string
	sampleManifestFile string
	unstructured       *unstructured.Unstructured
	workload           workload.Workload
	collectionTester   *E2ETest
	children           []client.Object
	getChildrenFunc    getChildren
	logSyntax          string
}

type getChildren func(*E2ETest) error
type readyChecker func() (bool, error)

// TestMain drives the whole suite: collections first (serial), then
// components (parallel), then teardown and a controller log scan.
func TestMain(t *testing.T) {
	e2eTestSuite := new(E2ETestSuiteConfig)
	require.NoErrorf(t, setupSuite(e2eTestSuite), "error setting up test suite")

	collectionSuite := &E2ECollectionTestSuite{suiteConfig: *e2eTestSuite}
	componentSuite := &E2EComponentTestSuite{suiteConfig: *e2eTestSuite}

	t.Run("TestE2ESuite", func(t *testing.T) {
		suite.Run(t, collectionSuite)
		suite.Run(t, componentSuite)
	})

	componentSuite.teardown()
	collectionSuite.teardown()

	if os.Getenv("DEPLOY_IN_CLUSTER") == "true" {
		require.NoErrorf(t, testControllerLogsNoErrors(e2eTestSuite, ""), "found errors in controller logs")
	}

	require.NoErrorf(t, finalTeardown(), "error tearing down test suite")
}

// setupSuite builds the clients from KUBECONFIG (or ~/.kube/config),
// loads the controller kustomization metadata, and deploys if requested.
func setupSuite(s *E2ETestSuiteConfig) error {
	var err error

	var config *rest.Config

	if os.Getenv("KUBECONFIG") != "" {
		config, err = clientcmd.BuildConfigFromFlags("", os.Getenv("KUBECONFIG"))
	} else {
		config, err = clientcmd.BuildConfigFromFlags("", os.Getenv("HOME")+"/.kube/config")
	}

	if err != nil {
		return fmt.Errorf("unable to create rest config from kubeconfig; %w", err)
	}

	restClient, err := kubernetes.NewForConfig(config)
	if err != nil {
		return fmt.Errorf("unable create rest client from kubeconfig; %w", err)
	}

	s.client = *restClient

	s.dynamicClient, err = dynamic.NewForConfig(config)
	if err != nil {
		return fmt.Errorf("unable to create dynamic client from kubeconfig; %w", err)
	}

	if err := readYamlFile(controllerKustomization, &s.controllerConfig); err != nil {
		return fmt.Errorf("unable to fetch controller configuration; %w", err)
	}

	return deploy(s)
}

// SetupTest marks every component test parallel.
func (s *E2EComponentTestSuite) SetupTest() {
	s.T().Parallel()
}

// setup loads the sample manifest, aligns namespaces, resolves children
// and creates the per-test namespace.
func (tester *E2ETest) setup() error {
	yamlFile, err := readYamlManifest(tester.sampleManifestFile, tester.unstructured)
	if err != nil {
		return fmt.Errorf("unable to fetch sample manifest; %w", err)
	}

	if err := k8syaml.Unmarshal(yamlFile, tester.workload); err != nil {
		return fmt.Errorf("unable to unmarshal yaml to api object; %w", err)
	}

	tester.unstructured.SetNamespace(tester.namespace)
	tester.workload.SetNamespace(tester.namespace)

	if tester.collectionTester != nil {
		collection := &unstructured.Unstructured{}

		collectionYaml, err := readYamlManifest(tester.collectionTester.sampleManifestFile, collection)
		if err != nil {
			return fmt.Errorf("unable to fetch sample collection manifest; %w", err)
		}

		if err := k8syaml.Unmarshal(collectionYaml, tester.collectionTester.workload); err != nil {
			return fmt.Errorf("unable to unmarshal collection yaml to api object; %w", err)
		}

		tester.collectionTester.unstructured.SetNamespace(tester.collectionTester.namespace)
		tester.collectionTester.workload.SetNamespace(tester.collectionTester.namespace)
	}

	if err := tester.getChildrenFunc(tester); err != nil {
		return fmt.Errorf("unable to resolve child resources; %w", err)
	}

	// cluster-scoped testers have no namespace and skip namespace creation
	if tester.namespace != "" {
		if err := createNamespaceForTest(tester); err != nil {
			return fmt.Errorf("failed to create namespace for test; %w", err)
		}
	}

	return nil
}

// deploy optionally installs CRDs and the controller:
//   DEPLOY="true"            -> make install (+ docker-build/push + deploy)
//   DEPLOY_IN_CLUSTER="true" -> wait for the in-cluster controller
func deploy(s *E2ETestSuiteConfig) error {
	if os.Getenv("DEPLOY") == "true" {
		if _, err := exec.Command("make", "-C", "../..", "install").Output(); err != nil {
			return fmt.Errorf("failed to run 'make install' target; %w", err)
		}
	}

	if os.Getenv("DEPLOY_IN_CLUSTER") == "true" {
		if os.Getenv("DEPLOY") == "true" {
			for _, target := range []string{"docker-build", "docker-push", "deploy"} {
				if _, err := exec.Command("make", "-C", "../..", target).Output(); err != nil {
					return fmt.Errorf("failed to run 'make %s' target; %w", target, err)
				}
			}
		}

		if err := waitForController(s); err != nil {
			return fmt.Errorf("failed to wait for controller for test; %w", err)
		}
	}

	return nil
}

// finalTeardown undeploys/uninstalls when TEARDOWN="true".
func finalTeardown() error {
	if os.Getenv("TEARDOWN") == "true" {
		target := "uninstall"
		if os.Getenv("DEPLOY_IN_CLUSTER") == "true" {
			target = "undeploy"
		}

		if _, err := exec.Command("make", "-C", "../..", target).Output(); err != nil {
			return fmt.Errorf("failed to run 'make %s' target with error; %w", target, err)
		}
	}

	return nil
}

func teardownSuite(s *E2ETestSuiteConfig) error {
	for _, e2eTest := range s.tests {
		if err := deleteCustomResource(e2eTest); err != nil {
			return fmt.Errorf("failed to delete custom resource: %+v; %w", e2eTest, err)
		}

		if e2eTest.namespace != "" {
			if err := deleteNamespaceForTest(e2eTest); err != nil {
				return fmt.Errorf("failed to delete namespace during teardown: %s; %w", e2eTest.namespace, err)
			}
		}
	}

	return nil
}

func (s *E2ECollectionTestSuite) teardown() {
	if len(s.suiteConfig.tests) > 0 {
		require.NoErrorf(s.T(), teardownSuite(&s.suiteConfig), "unable to teardown collection test suite")
	}
}

func (s *E2EComponentTestSuite) teardown() {
	if len(s.suiteConfig.tests) > 0 {
		require.NoErrorf(s.T(), teardownSuite(&s.suiteConfig), "unable to teardown component test suite")
	}
}

//
// manifest + client helpers
//

func readYamlManifest(path string, destination *unstructured.Unstructured) ([]byte, error) {
	yamlFile, err := os.ReadFile(path)
	if err != nil {
		return nil, fmt.Errorf("unable to read file %s; %w", path, err)
	}

	dec := serializer.NewDecodingSerializer(unstructured.UnstructuredJSONScheme)

	if _, _, err = dec.Decode(yamlFile, nil, destination); err != nil {
		return nil, fmt.Errorf("error decoding sample manifest %s; %w\n\nwith data: %s", path, err, yamlFile)
	}

	return yamlFile, nil
}

func readYamlFile(path string, destination interface{}) error {
	yamlFile, err := os.ReadFile(path)
	if err != nil {
		return fmt.Errorf("unable to read file %s; %w", path, err)
	}

	if err = yaml.Unmarshal(yamlFile, destination); err != nil {
		return fmt.Errorf("unable to unmarshal yaml file %s; %w", path, err)
	}

	return nil
}

func namespaceExists(tester *E2ETest) (bool, error) {
	_, err := tester.suiteConfig.client.CoreV1().Namespaces().Get(
		context.TODO(),
		tester.namespace,
		metav1.GetOptions{},
	)
	if err != nil {
		if errors.IsNotFound(err) {
			return false, nil
		}

		return false, err
	}

	return true, nil
}

func getPlural(kind string) string {
	pluralMap := map[string]string{
		"resourcequota": "resourcequotas",
	}

	plural := kbresource.RegularPlural(kind)

	if pluralMap[plural] != "" {
		return pluralMap[plural]
	}

	return plural
}

func getDeletableChild(tester *E2ETest) client.Object {
	for _, whitelistKind := range deletableWhitelist {
		for _, child := range tester.children {
			if child.GetObjectKind().GroupVersionKind().Kind == whitelistKind {
				return child
			}
		}
	}

	return nil
}

func getResourceGVR(resource client.Object) schema.GroupVersionResource {
	return schema.GroupVersionResource{
		Group:    resource.GetObjectKind().GroupVersionKind().Group,
		Version:  resource.GetObjectKind().GroupVersionKind().Version,
		Resource: getPlural(strings.ToLower(resource.GetObjectKind().GroupVersionKind().Kind)),
	}
}

func getClientForResource(tester *E2ETest, resource client.Object) dynamic.ResourceInterface {
	if tester.namespace != "" {
		return tester.suiteConfig.dynamicClient.Resource(getResourceGVR(resource)).
			Namespace(tester.namespace)
	}

	return tester.suiteConfig.dynamicClient.Resource(getResourceGVR(resource)).
		Namespace(resource.GetNamespace())
}

func getControllerDeployment(s *E2ETestSuiteConfig) (*appsv1.Deployment, error) {
	return s.client.
		AppsV1().Deployments(s.controllerConfig.Namespace).
		Get(context.TODO(), (s.controllerConfig.Prefix + controllerName), metav1.GetOptions{})
}

func createNamespaceForTest(tester *E2ETest) error {
	exists, err := namespaceExists(tester)
	if exists || err != nil {
		return err
	}

	namespace := &v1.Namespace{
		TypeMeta: metav1.TypeMeta{
			APIVersion: resources.NamespaceVersion,
			Kind:       resources.NamespaceKind,
		},
		ObjectMeta: metav1.ObjectMeta{
			Name: tester.namespace,
		},
	}

	_, err = tester.suiteConfig.client.
		CoreV1().Namespaces().
		Create(context.TODO(), namespace, metav1.CreateOptions{})

	return err
}

func getControllerLogs(s *E2ETestSuiteConfig) (string, error) {
	deployment, err := getControllerDeployment(s)
	if err != nil {
		return "", fmt.Errorf("unable to retrieve controller deployment; %w", err)
	}

	podListOpts := metav1.ListOptions{
		LabelSelector: labels.SelectorFromSet(deployment.Spec.Template.Labels).String(),
	}

	controllerPods, err := s.client.CoreV1().Pods(s.controllerConfig.Namespace).List(context.TODO(), podListOpts)
	if err != nil {
		return "", fmt.Errorf("unable to retrieve controller pods; %w", err)
	}

	buf := new(bytes.Buffer)

	for _, pod := range controllerPods.Items {
		for _, container := range pod.Spec.Containers {
			podLogOpts := v1.PodLogOptions{Container: container.Name}
			req := s.client.CoreV1().Pods(pod.Namespace).GetLogs(pod.Name, &podLogOpts)

			podLogs, err := req.Stream(context.TODO())
			if err != nil {
				return "", fmt.Errorf("error opening log stream for pod %s/%s; %w", pod.Namespace, pod.Name, err)
			}

			defer podLogs.Close()

			if _, err := io.Copy(buf, podLogs); err != nil {
				return "", fmt.Errorf("error storing logs to string buffer; %w", err)
			}
		}
	}

	return buf.String(), nil
}

func updateResource(tester *E2ETest, resource client.Object) error {
	unstructuredResource, err := resources.ToUnstructured(resource)
	if err != nil {
		return err
	}

	_, err = getClientForResource(tester, resource).
		Update(context.TODO(), unstructuredResource, metav1.UpdateOptions{})

	return err
}

func deleteResource(tester *E2ETest, resource client.Object) error {
	return getClientForResource(tester, resource).
		Delete(context.TODO(), resource.GetName(), metav1.DeleteOptions{})
}

func deleteCustomResource(tester *E2ETest) error {
	crClient := getClientForResource(tester, tester.unstructured)

	_, err := crClient.Get(context.TODO(), tester.unstructured.GetName(), metav1.GetOptions{})
	if err != nil {
		if errors.IsNotFound(err) {
			return nil
		}

		return err
	}

	if err := crClient.Delete(context.TODO(), tester.unstructured.GetName(), metav1.DeleteOptions{}); err != nil {
		return fmt.Errorf("error deleting custom resource: %+v; %w", tester.unstructured, err)
	}

	return waitForMissingResources(tester)
}

func deleteNamespaceForTest(tester *E2ETest) error {
	err := tester.suiteConfig.client.
		CoreV1().Namespaces().
		Delete(context.TODO(), tester.namespace, metav1.DeleteOptions{})
	if err != nil {
		return err
	}

	namespaceIsMissing := func() (bool, error) {
		exists, err := namespaceExists(tester)
		if err != nil {
			return false, err
		}

		return !exists, nil
	}

	return waitFor(namespaceIsMissing)
}

//
// wait helpers
//

func waitForMissingResources(tester *E2ETest) error {
	childResourcesAreMissing := func() (bool, error) {
		for _, child := range tester.children {
			_, err := getClientForResource(tester, child).
				Get(context.TODO(), child.GetName(), metav1.GetOptions{})

			if err == nil {
				return false, nil
			}

			if errors.IsNotFound(err) {
				continue
			}

			return false, err
		}

		return true, nil
	}

	return waitFor(childResourcesAreMissing)
}

func waitForChildResources(tester *E2ETest) error {
	childResourcesAreReady := func() (bool, error) {
		childResourceClusterObjects := make([]client.Object, len(tester.children))

		for i, child := range tester.children {
			childResourceClusterObject, err := getClientForResource(tester, child).
				Get(context.TODO(), child.GetName(), metav1.GetOptions{})
			if err != nil {
				return false, fmt.Errorf("unable to get child resource from cluster: %+v; %w", child, err)
			}

			childResourceClusterObjects[i] = childResourceClusterObject
		}

		return resources.AreReady(childResourceClusterObjects...)
	}

	return waitFor(childResourcesAreReady)
}

func waitForCustomResource(tester *E2ETest) error {
	customResourceIsReady := func() (bool, error) {
		customResource, err := getClientForResource(tester, tester.unstructured).
			Get(context.TODO(), tester.unstructured.GetName(), metav1.GetOptions{})
		if err != nil {
			return false, fmt.Errorf("unable to get custom resource from cluster: %+v; %w", customResource, err)
		}

		if customResource.Object["status"] == nil {
			return false, nil
		}

		createStatus := customResource.Object["status"].(map[string]interface{})["created"]
		if createStatus != nil {
			created, ok := createStatus.(bool)
			if !ok {
				return false, fmt.Errorf("unable to determine custom resource status")
			}

			return created, nil
		}

		return false, nil
	}

	return waitFor(customResourceIsReady)
}

func waitForController(s *E2ETestSuiteConfig) error {
	deploymentIsReady := func() (bool, error) {
		deployment, err := getControllerDeployment(s)
		if err != nil {
			return false, err
		}

		return resources.IsReady(deployment)
	}

	return waitFor(deploymentIsReady)
}

func waitFor(isReady readyChecker) error {
	timeout, interval := time.After(waitTimeout), time.Tick(waitInterval)

	for {
		select {
		case <-timeout:
			return fmt.Errorf("timed out waiting for resource")
		case <-interval:
			ready, err := isReady()
			if err != nil {
				return fmt.Errorf("error waiting for resource to be ready, %w", err)
			}

			if ready {
				return nil
			}
		}
	}
}

//
// test bodies shared by every workload test file
//

func testCreateCustomResource(tester *E2ETest) error {
	_, err := getClientForResource(tester, tester.unstructured).
		Create(context.TODO(), tester.unstructured, metav1.CreateOptions{})
	if err != nil {
		return fmt.Errorf("error creating custom resource: %+v; %w", tester.unstructured, err)
	}

	if err = waitForCustomResource(tester); err != nil {
		return fmt.Errorf("failed waiting for custom resource ready status: %v; %w", tester.unstructured, err)
	}

	if err = waitForChildResources(tester); err != nil {
		return fmt.Errorf("child resources are not in a ready state: %v; %w", tester.unstructured, err)
	}

	return nil
}

func testDeleteChildResource(tester *E2ETest) error {
	childToDelete := getDeletableChild(tester)
	if childToDelete != nil {
		if err := deleteResource(tester, childToDelete); err != nil {
			return fmt.Errorf("failed deleting child resource;: %+v; %w", childToDelete, err)
		}

		if err := waitForChildResources(tester); err != nil {
			return fmt.Errorf(
				"failed waiting for reconciliation after child deletion for resource: %+v; %w",
				childToDelete,
				err,
			)
		}
	}

	return nil
}

func testControllerLogsNoErrors(s *E2ETestSuiteConfig, searchSyntax string) error {
	logs, err := getControllerLogs(s)
	if err != nil {
		return fmt.Errorf("failed fetching controller logs; %w", err)
	}

	foundErrors := []string{}

	for _, logLine := range strings.Split(logs, "\n") {
		if strings.Contains(logLine, "ERROR") && strings.Contains(logLine, searchSyntax) {
			foundErrors = append(foundErrors, logLine)
		}
	}

	if len(foundErrors) > 0 {
		return fmt.Errorf("found errors in controller: +%v", foundErrors)
	}

	return nil
}
