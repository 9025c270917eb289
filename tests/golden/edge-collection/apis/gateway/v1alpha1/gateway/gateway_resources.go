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

package gateway

import (
	"k8s.io/apimachinery/pkg/apis/meta/v1/unstructured"
	"sigs.k8s.io/controller-runtime/pkg/client"

	edgeplatformv1alpha1 "github.com/acme/edgeplatform/apis/edgeplatform/v1alpha1"
	gatewayv1alpha1 "github.com/acme/edgeplatform/apis/gateway/v1alpha1"
)

// +kubebuilder:rbac:groups=apps,resources=deployments,verbs=get;list;watch;create;update;patch;delete

const DeploymentEdgeSystemEdgeGateway = "edge-gateway"

// CreateDeploymentEdgeSystemEdgeGateway creates the edge-gateway Deployment resource.
func CreateDeploymentEdgeSystemEdgeGateway(
	parent *gatewayv1alpha1.EdgeGateway,
	collection *edgeplatformv1alpha1.EdgePlatform,
) ([]client.Object, error) {
	if collection.Spec.Region != "us-east-1" {
		return []client.Object{}, nil
	}

	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "apps/v1",
			"kind": "Deployment",
			"metadata": map[string]interface{}{
				"name": "edge-gateway",
				"namespace": "edge-system",
			},
			"spec": map[string]interface{}{
				"replicas": parent.Spec.GatewayReplicas,
				"selector": map[string]interface{}{
					"matchLabels": map[string]interface{}{
						"app": "edge-gateway",
					},
				},
				"template": map[string]interface{}{
					"metadata": map[string]interface{}{
						"labels": map[string]interface{}{
							"app": "edge-gateway",
						},
					},
					"spec": map[string]interface{}{
						"containers": []interface{}{
							map[string]interface{}{
								"name": "gateway",
								"image": collection.Spec.GatewayImage,
							},
						},
					},
				},
			},
		},
	}

	resourceObj.SetNamespace(parent.Namespace)

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}

// +kubebuilder:rbac:groups=core,resources=services,verbs=get;list;watch;create;update;patch;delete

const ServiceEdgeSystemEdgeGatewaySvc = "edge-gateway-svc"

// CreateServiceEdgeSystemEdgeGatewaySvc creates the edge-gateway-svc Service resource.
func CreateServiceEdgeSystemEdgeGatewaySvc(
	parent *gatewayv1alpha1.EdgeGateway,
	collection *edgeplatformv1alpha1.EdgePlatform,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "v1",
			"kind": "Service",
			"metadata": map[string]interface{}{
				"name": "edge-gateway-svc",
				"namespace": "edge-system",
			},
			"spec": map[string]interface{}{
				"selector": map[string]interface{}{
					"app": "edge-gateway",
				},
				"ports": []interface{}{
					map[string]interface{}{
						"port": 443,
						"targetPort": 8443,
					},
				},
			},
		},
	}

	resourceObj.SetNamespace(parent.Namespace)

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}
