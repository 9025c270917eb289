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

package mesh

import (
	"fmt"

	"sigs.k8s.io/controller-runtime/pkg/client"
	"sigs.k8s.io/yaml"

	"github.com/nukleros/operator-builder-tools/pkg/controller/workload"

	edgeplatformv1alpha1 "github.com/acme/edgeplatform/apis/edgeplatform/v1alpha1"
	meshv1alpha1 "github.com/acme/edgeplatform/apis/mesh/v1alpha1"
)

// sampleServiceMesh is a sample containing all fields
const sampleServiceMesh = `apiVersion: mesh.example.com/v1alpha1
kind: ServiceMesh
metadata:
  name: servicemesh-sample
spec:
  #collection:
    #name: "edgeplatform-sample"
    #namespace: ""
  meshAgentImage: "meshd:v0.9"
`

// sampleServiceMeshRequired is a sample containing only required fields
const sampleServiceMeshRequired = `apiVersion: mesh.example.com/v1alpha1
kind: ServiceMesh
metadata:
  name: servicemesh-sample
spec:
  #collection:
    #name: "edgeplatform-sample"
    #namespace: ""
`

// Sample returns the sample manifest for this custom resource.
func Sample(requiredOnly bool) string {
	if requiredOnly {
		return sampleServiceMeshRequired
	}

	return sampleServiceMesh
}

// Generate returns the child resources that are associated with this workload given
// appropriate structured inputs.
func Generate(
	workloadObj meshv1alpha1.ServiceMesh,
	collectionObj edgeplatformv1alpha1.EdgePlatform,
) ([]client.Object, error) {
	resourceObjects := []client.Object{}

	for _, f := range CreateFuncs {
		resources, err := f(&workloadObj, &collectionObj)

		if err != nil {
			return nil, err
		}

		resourceObjects = append(resourceObjects, resources...)
	}

	return resourceObjects, nil
}

// GenerateForCLI returns the child resources that are associated with this workload given
// appropriate YAML manifest files.
func GenerateForCLI(workloadFile []byte, collectionFile []byte) ([]client.Object, error) {
	var workloadObj meshv1alpha1.ServiceMesh
	if err := yaml.Unmarshal(workloadFile, &workloadObj); err != nil {
		return nil, fmt.Errorf("failed to unmarshal yaml into workload, %w", err)
	}

	if err := workload.Validate(&workloadObj); err != nil {
		return nil, fmt.Errorf("error validating workload yaml, %w", err)
	}

	var collectionObj edgeplatformv1alpha1.EdgePlatform
	if err := yaml.Unmarshal(collectionFile, &collectionObj); err != nil {
		return nil, fmt.Errorf("failed to unmarshal yaml into collection, %w", err)
	}

	if err := workload.Validate(&collectionObj); err != nil {
		return nil, fmt.Errorf("error validating collection yaml, %w", err)
	}

	return Generate(workloadObj, collectionObj)
}

// CreateFuncs is an array of functions that are called to create the child resources for the controller
// in memory during the reconciliation loop prior to persisting the changes or updates to the Kubernetes
// database.
var CreateFuncs = []func(
	*meshv1alpha1.ServiceMesh,
	*edgeplatformv1alpha1.EdgePlatform,
) ([]client.Object, error) {
	CreateClusterRoleMeshAgent,
	CreateDaemonSetEdgeSystemMeshAgent,
	CreateCustomResourceDefinitionMeshpoliciesMeshExampleCom,
}

// InitFuncs is an array of functions that are called prior to starting the controller manager.  This is
// necessary in instances which the controller needs to "own" objects which depend on resources to
// pre-exist in the cluster. A common use case for this is the need to own a custom resource.
// If the controller needs to own a custom resource type, the CRD that defines it must
// first exist. In this case, the InitFunc will create the CRD so that the controller
// can own custom resources of that type.  Without the InitFunc the controller will
// crash loop because when it tries to own a non-existent resource type during manager
// setup, it will fail.
var InitFuncs = []func(
	*meshv1alpha1.ServiceMesh,
	*edgeplatformv1alpha1.EdgePlatform,
) ([]client.Object, error) {
	CreateCustomResourceDefinitionMeshpoliciesMeshExampleCom,
}

func ConvertWorkload(component, collection workload.Workload) (
	*meshv1alpha1.ServiceMesh,
	*edgeplatformv1alpha1.EdgePlatform,
	error,
) {
	p, ok := component.(*meshv1alpha1.ServiceMesh)
	if !ok {
		return nil, nil, meshv1alpha1.ErrUnableToConvertServiceMesh
	}

	c, ok := collection.(*edgeplatformv1alpha1.EdgePlatform)
	if !ok {
		return nil, nil, edgeplatformv1alpha1.ErrUnableToConvertEdgePlatform
	}

	return p, c, nil
}
